"""Sentinel error contract mapped across the RPC boundary.

Mirrors the reference's internal/errdefs + pkg/api/kukeonv1/errmap pattern:
each error type has a stable wire code; the RPC client reconstructs the
typed error from the code so `except CellNotFound:` works on both sides of
the socket.
"""
from __future__ import annotations


class KukeonError(Exception):
    code = "ErrUnknown"

    def __init__(self, msg: str = ""):
        super().__init__(msg or self.__doc__ or self.code)


class NotFound(KukeonError):
    code = "ErrNotFound"


class RealmNotFound(NotFound):
    code = "ErrRealmNotFound"


class SpaceNotFound(NotFound):
    code = "ErrSpaceNotFound"


class StackNotFound(NotFound):
    code = "ErrStackNotFound"


class CellNotFound(NotFound):
    code = "ErrCellNotFound"


class ContainerNotFound(NotFound):
    code = "ErrContainerNotFound"


class SessionNotFound(NotFound):
    code = "ErrSessionNotFound"


class SecretNotFound(NotFound):
    code = "ErrSecretNotFound"


class BlueprintNotFound(NotFound):
    code = "ErrBlueprintNotFound"


class ConfigNotFound(NotFound):
    code = "ErrConfigNotFound"


class VolumeNotFound(NotFound):
    code = "ErrVolumeNotFound"


class AlreadyExists(KukeonError):
    code = "ErrAlreadyExists"


class Conflict(KukeonError):
    code = "ErrConflict"


class InvalidArgument(KukeonError):
    code = "ErrInvalidArgument"


class ValidationError(InvalidArgument):
    code = "ErrValidation"


class NotReady(KukeonError):
    code = "ErrNotReady"


class StaleResource(KukeonError):
    """Optimistic-concurrency conflict: the resource generation moved."""
    code = "ErrStaleResource"


class DiskPressure(KukeonError):
    code = "ErrDiskPressure"


class GPUUnavailable(KukeonError):
    """No free MI355X devices for the requested pin count."""
    code = "ErrGPUUnavailable"


class AttachPingTimeout(KukeonError):
    code = "ErrAttachPingTimeout"


class NotEmpty(KukeonError):
    """Scope still has children (delete without cascade)."""
    code = "ErrNotEmpty"


_BY_CODE = {}


def _walk(cls):
    _BY_CODE[cls.code] = cls
    for sub in cls.__subclasses__():
        _walk(sub)


_walk(KukeonError)


def to_wire(err: Exception) -> dict:
    code = getattr(err, "code", "ErrUnknown")
    return {"code": code, "message": str(err)}


def from_wire(data: dict) -> KukeonError:
    cls = _BY_CODE.get(data.get("code", ""), KukeonError)
    return cls(data.get("message", ""))
