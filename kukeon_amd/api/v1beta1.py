"""v1beta1 API model: external YAML/JSON document types for every kind.

Mirrors the reference's manifest contract (field names, kinds, state strings
— pkg/api/model/v1beta1: kinds consts.go:24-84, CellSpec cell.go:45-118,
ContainerSpec container.go:34-181, restart policy container.go:126-141,
state enums cell.go:242-271, SpaceSpec egress space.go:38-79; studied, not
copied) with two first-class
additions from the agent-native proposal: the Session and Interactive kinds
(docs/site/proposals/agent-native-orchestration.md:179-211), plus the GPU
fields the MI355X runtime needs (ContainerSpec.gpus → amdgpu device pinning
+ ROCR_VISIBLE_DEVICES).

Serialization: snake_case Python attributes <-> camelCase YAML/JSON keys,
omitting empty optionals (the reference's `omitempty` contract) so docs
round-trip cleanly through metadata.json and `kuke get -o yaml`.
"""
from __future__ import annotations

import copy
import dataclasses
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, get_args, get_origin

API_VERSION = "v1beta1"

KIND_REALM = "Realm"
KIND_SPACE = "Space"
KIND_STACK = "Stack"
KIND_CELL = "Cell"
KIND_CONTAINER = "Container"
KIND_SESSION = "Session"
KIND_SECRET = "Secret"
KIND_CELL_BLUEPRINT = "CellBlueprint"
KIND_CELL_CONFIG = "CellConfig"
KIND_VOLUME = "Volume"
KIND_SERVER_CONFIGURATION = "ServerConfiguration"
KIND_CLIENT_CONFIGURATION = "ClientConfiguration"

ALL_KINDS = [
    KIND_REALM, KIND_SPACE, KIND_STACK, KIND_CELL, KIND_CONTAINER,
    KIND_SESSION, KIND_SECRET, KIND_CELL_BLUEPRINT, KIND_CELL_CONFIG,
    KIND_VOLUME, KIND_SERVER_CONFIGURATION, KIND_CLIENT_CONFIGURATION,
]

# Dependency order for `kuke apply` (reference: internal/controller/apply.go)
APPLY_ORDER = [
    KIND_REALM, KIND_SPACE, KIND_STACK, KIND_SECRET, KIND_VOLUME,
    KIND_CELL_BLUEPRINT, KIND_CELL_CONFIG, KIND_CELL, KIND_SESSION,
]

LABEL_TEAM = "kukeon.io/team"
LABEL_BLUEPRINT = "kukeon.io/blueprint"
ANNOTATION_SOURCE_CELL = "kukeon.io/source-cell"

# ---- state strings (ordinal-stable contract) ----
STATE_PENDING = "Pending"
STATE_READY = "Ready"
STATE_STOPPED = "Stopped"
STATE_PAUSED = "Paused"
STATE_FAILED = "Failed"
STATE_UNKNOWN = "Unknown"
STATE_CREATING = "Creating"
STATE_DELETING = "Deleting"
STATE_NOT_CREATED = "NotCreated"
STATE_EXITED = "Exited"       # clean self-exit terminal (all workloads rc 0)
STATE_ERROR = "Error"         # workload-crash terminal (some workload rc != 0)
STATE_DEGRADED = "Degraded"   # partial health, non-terminal
STATE_RUNNING = "Running"     # session
STATE_COMPLETED = "Completed"  # session terminal (clean end)
STATE_TERMINATED = "Terminated"  # session terminal (deadline/explicit kill)

TERMINAL_CELL_STATES = {STATE_FAILED, STATE_ERROR, STATE_EXITED}

RESTART_NEVER = ""
RESTART_ALWAYS = "always"
RESTART_ON_FAILURE = "on-failure"
DEFAULT_RESTART_BACKOFF_SECONDS = 30
DEFAULT_RESTART_MAX_RETRIES = 5


# ---------------------------------------------------------------------------
# serde framework
# ---------------------------------------------------------------------------
def _camel(name: str) -> str:
    parts = name.split("_")
    out = parts[0] + "".join(p[:1].upper() + p[1:] for p in parts[1:])
    # acronym fixes matching the reference's JSON tags
    for a, b in (("Id", "ID"), ("Cni", "CNI"), ("Gid", "GID"),
                 ("Pid", "PID")):
        if out.endswith(a):
            out = out[: -len(a)] + b
    return out


_KEY_OVERRIDES = {
    "realm_id": "realmId", "space_id": "spaceId", "stack_id": "stackId",
    "cell_id": "cellId", "id": "id", "host_pid": "hostPID",
    "cni_config_path": "cniConfigPath",
    "kukeon_group_gid": "kukeonGroupGID",
}


def _json_key(f: dataclasses.Field) -> str:
    return f.metadata.get("key") or _KEY_OVERRIDES.get(f.name) or _camel(f.name)


def _is_doc(tp) -> bool:
    return dataclasses.is_dataclass(tp)


def _to_dict(obj) -> Any:
    if dataclasses.is_dataclass(obj):
        out = {}
        for f in dataclasses.fields(obj):
            v = getattr(obj, f.name)
            omit = f.metadata.get("omitempty", True)
            if v is None:
                continue
            dv = _to_dict(v)
            if omit and (dv == {} or dv == [] or dv == "" or dv is None):
                # keep explicit False/0 only when not omitempty
                if dv in ({}, [], "", None):
                    continue
            if omit and dv is False:
                continue
            if omit and dv == 0 and isinstance(dv, int) and not isinstance(dv, bool) \
                    and f.metadata.get("omitzero", False):
                continue
            out[_json_key(f)] = dv
        return out
    if isinstance(obj, list):
        return [_to_dict(x) for x in obj]
    if isinstance(obj, dict):
        return {k: _to_dict(v) for k, v in obj.items()}
    return obj


def _from_dict(cls, data):
    if data is None:
        return None
    if not dataclasses.is_dataclass(cls):
        return data
    if not isinstance(data, dict):
        raise ValueError(
            f"{cls.__name__}: expected a mapping, got "
            f"{type(data).__name__} ({data!r})")
    kwargs = {}
    for f in dataclasses.fields(cls):
        key = _json_key(f)
        if key not in data:
            continue
        v = data[key]
        tp = f.type
        # resolve typing constructs
        origin = get_origin(tp) if not isinstance(tp, str) else None
        if isinstance(tp, str):
            tp = _TYPE_REGISTRY.get(tp.replace("Optional[", "").rstrip("]"),
                                    None) or tp
        if dataclasses.is_dataclass(tp):
            kwargs[f.name] = _from_dict(tp, v)
        elif origin is list and v is not None:
            (elem,) = get_args(f.type)
            if dataclasses.is_dataclass(elem):
                kwargs[f.name] = [_from_dict(elem, x) for x in v]
            else:
                kwargs[f.name] = list(v)
        else:
            # string annotations (from __future__): use registry lookup
            elem = _elem_type(f)
            if elem is not None and isinstance(v, list):
                kwargs[f.name] = [
                    _from_dict(elem, x) if dataclasses.is_dataclass(elem)
                    else x for x in v]
            elif elem is not None and isinstance(v, dict) and \
                    dataclasses.is_dataclass(elem):
                kwargs[f.name] = _from_dict(elem, v)
            else:
                kwargs[f.name] = v
    return cls(**kwargs)


_TYPE_REGISTRY: Dict[str, type] = {}


def _elem_type(f: dataclasses.Field):
    """Resolve dataclass element type from string annotations like
    'List[ContainerSpec]' or 'Optional[CellTty]'."""
    t = f.type if isinstance(f.type, str) else None
    if t is None:
        return None
    t = t.replace("Optional[", "").replace("List[", "").rstrip("]")
    return _TYPE_REGISTRY.get(t)


def register(cls):
    _TYPE_REGISTRY[cls.__name__] = cls
    return cls


class DocBase:
    def to_dict(self) -> Dict[str, Any]:
        return _to_dict(self)

    @classmethod
    def from_dict(cls, data: Dict[str, Any]):
        return _from_dict(cls, data)

    def deep_copy(self):
        return copy.deepcopy(self)


# ---------------------------------------------------------------------------
# shared metadata
# ---------------------------------------------------------------------------
@register
@dataclass
class Metadata(DocBase):
    name: str = ""
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    generation: int = field(default=0, metadata={"omitzero": True})


# ---------------------------------------------------------------------------
# Realm / Space / Stack
# ---------------------------------------------------------------------------
@register
@dataclass
class RealmSpec(DocBase):
    description: str = ""


@register
@dataclass
class ScopeStatus(DocBase):
    state: str = field(default=STATE_PENDING, metadata={"omitempty": False})
    cgroup_path: str = ""
    observed_generation: int = field(default=0, metadata={"omitzero": True})


@register
@dataclass
class RealmDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_REALM, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: RealmSpec = field(default_factory=RealmSpec)
    status: ScopeStatus = field(default_factory=ScopeStatus)


@register
@dataclass
class EgressAllowRule(DocBase):
    host: str = ""
    cidr: str = ""
    ports: List[int] = field(default_factory=list)


@register
@dataclass
class EgressPolicy(DocBase):
    default: str = field(default="allow", metadata={"omitempty": False})
    allow: List[EgressAllowRule] = field(default_factory=list)


@register
@dataclass
class SpaceNetwork(DocBase):
    egress: Optional[EgressPolicy] = None


@register
@dataclass
class ContainerResources(DocBase):
    memory_limit_bytes: int = field(default=0, metadata={"omitzero": True})
    cpu_shares: int = field(default=0, metadata={"omitzero": True})


@register
@dataclass
class ContainerCapabilities(DocBase):
    add: List[str] = field(default_factory=list)
    drop: List[str] = field(default_factory=list)


@register
@dataclass
class SpaceContainerDefaults(DocBase):
    user: str = ""
    read_only_root_filesystem: Optional[bool] = None
    capabilities: Optional[ContainerCapabilities] = None
    security_opts: List[str] = field(default_factory=list)
    resources: Optional[ContainerResources] = None


@register
@dataclass
class SpaceDefaults(DocBase):
    container: Optional[SpaceContainerDefaults] = None


@register
@dataclass
class SpaceSpec(DocBase):
    realm_id: str = field(default="", metadata={"omitempty": False})
    cni_config_path: str = ""
    # `network: {}` declares a networked space even with no egress rules
    # (bridge + per-cell netns); it must survive the wire round-trip, so
    # the empty object is not omitted
    network: Optional[SpaceNetwork] = field(
        default=None, metadata={"omitempty": False})
    defaults: Optional[SpaceDefaults] = None


@register
@dataclass
class SpaceStatus(DocBase):
    state: str = field(default=STATE_PENDING, metadata={"omitempty": False})
    cgroup_path: str = ""
    bridge_name: str = ""
    subnet: str = ""
    observed_generation: int = field(default=0, metadata={"omitzero": True})


@register
@dataclass
class SpaceDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_SPACE, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: SpaceSpec = field(default_factory=SpaceSpec)
    status: SpaceStatus = field(default_factory=SpaceStatus)


@register
@dataclass
class StackSpec(DocBase):
    realm_id: str = field(default="", metadata={"omitempty": False})
    space_id: str = field(default="", metadata={"omitempty": False})
    description: str = ""


@register
@dataclass
class StackDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_STACK, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: StackSpec = field(default_factory=StackSpec)
    status: ScopeStatus = field(default_factory=ScopeStatus)


# ---------------------------------------------------------------------------
# Container
# ---------------------------------------------------------------------------
@register
@dataclass
class VolumeMount(DocBase):
    name: str = ""
    source: str = ""
    target: str = ""
    read_only: bool = False


@register
@dataclass
class ContainerSecret(DocBase):
    name: str = ""
    env: str = ""      # inject as env var
    path: str = ""     # or mount as file


@register
@dataclass
class ContainerRepo(DocBase):
    url: str = ""
    path: str = ""
    ref: str = ""


@register
@dataclass
class ContainerGit(DocBase):
    name: str = ""
    email: str = ""
    signing_key: str = ""


@register
@dataclass
class ContainerTty(DocBase):
    prompt: str = ""
    init_script: str = ""
    shell: str = ""


@register
@dataclass
class RepoStatus(DocBase):
    url: str = ""
    state: str = ""    # cloned | fetched | failed
    error: str = ""


@register
@dataclass
class ContainerSpec(DocBase):
    id: str = ""
    realm_id: str = ""
    space_id: str = ""
    stack_id: str = ""
    cell_id: str = ""
    root: bool = False
    image: str = field(default="", metadata={"omitempty": False})
    command: str = field(default="", metadata={"omitempty": False})
    args: List[str] = field(default_factory=list)
    working_dir: str = ""
    env: List[str] = field(default_factory=list)
    ports: List[str] = field(default_factory=list)
    volumes: List[VolumeMount] = field(default_factory=list)
    privileged: bool = False
    host_network: bool = False
    host_pid: bool = False
    user: str = ""
    read_only_root_filesystem: bool = False
    capabilities: Optional[ContainerCapabilities] = None
    security_opts: List[str] = field(default_factory=list)
    devices: List[str] = field(default_factory=list)
    # MI355X extension: number of GPUs to pin (amdgpu /dev/kfd +
    # /dev/dri/renderD* device access + ROCR_VISIBLE_DEVICES injection)
    gpus: int = field(default=0, metadata={"omitzero": True})
    resources: Optional[ContainerResources] = None
    secrets: List[ContainerSecret] = field(default_factory=list)
    repos: List[ContainerRepo] = field(default_factory=list)
    git: Optional[ContainerGit] = None
    restart_policy: str = field(default="", metadata={"omitempty": False})
    restart_backoff_seconds: Optional[int] = None
    restart_max_retries: Optional[int] = None
    attachable: bool = False
    tty: Optional[ContainerTty] = None


@register
@dataclass
class ContainerStatus(DocBase):
    state: str = field(default=STATE_PENDING, metadata={"omitempty": False})
    pid: int = field(default=0, metadata={"omitzero": True})
    exit_code: Optional[int] = None
    started_at: str = ""
    finished_at: str = ""
    restart_count: int = field(default=0, metadata={"omitzero": True})
    last_restart_at: str = ""
    repos: List[RepoStatus] = field(default_factory=list)
    gpu_ids: List[int] = field(default_factory=list)


@register
@dataclass
class ContainerDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_CONTAINER, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: ContainerSpec = field(default_factory=ContainerSpec)
    status: ContainerStatus = field(default_factory=ContainerStatus)


# ---------------------------------------------------------------------------
# Cell
# ---------------------------------------------------------------------------
@register
@dataclass
class CellProvenance(DocBase):
    binding_kind: str = ""     # "config" | "blueprint"
    binding_ref: str = ""      # scoped name of the binding
    params: Dict[str, str] = field(default_factory=dict)
    env: List[str] = field(default_factory=list)


@register
@dataclass
class CellTty(DocBase):
    prompt: str = ""
    init_script: str = ""


@register
@dataclass
class CellSpec(DocBase):
    id: str = ""
    realm_id: str = field(default="", metadata={"omitempty": False})
    space_id: str = field(default="", metadata={"omitempty": False})
    stack_id: str = field(default="", metadata={"omitempty": False})
    root_container_id: str = ""
    tty: Optional[CellTty] = None
    containers: List[ContainerSpec] = field(default_factory=list,
                                            metadata={"omitempty": False})
    auto_delete: bool = False
    runtime_env: List[str] = field(default_factory=list)
    provenance: Optional[CellProvenance] = None
    ignore_disk_pressure: bool = False


@register
@dataclass
class CellStatus(DocBase):
    state: str = field(default=STATE_PENDING, metadata={"omitempty": False})
    message: str = ""
    cgroup_path: str = ""
    observed_generation: int = field(default=0, metadata={"omitzero": True})
    out_of_sync: bool = False
    out_of_sync_reason: str = ""
    out_of_sync_error: str = ""
    containers: List[ContainerStatus] = field(default_factory=list)


@register
@dataclass
class CellDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_CELL, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: CellSpec = field(default_factory=CellSpec)
    status: CellStatus = field(default_factory=CellStatus)


# ---------------------------------------------------------------------------
# Session (agent-native lifetime primitive, proposal §4.5 made first-class)
# ---------------------------------------------------------------------------
@register
@dataclass
class SessionLifetime(DocBase):
    wall_clock: str = ""     # e.g. "30m"
    idle_timeout: str = ""   # e.g. "5m"


@register
@dataclass
class SessionPersist(DocBase):
    volume: str = ""


@register
@dataclass
class SessionOnEnd(DocBase):
    persist: List[SessionPersist] = field(default_factory=list)


@register
@dataclass
class SessionSpec(DocBase):
    realm_id: str = ""
    space_id: str = ""
    stack_id: str = field(default="", metadata={"omitempty": False})
    cell_id: str = ""        # the cell this session runs (ours: 1 cell/session)
    owner: str = ""
    task: str = ""
    # MI355X extension: GPUs pinned for the session's cell + the modelhub
    # endpoint that backs the agent
    gpus: int = field(default=0, metadata={"omitzero": True})
    modelhub: str = ""
    lifetime: Optional[SessionLifetime] = None
    on_end: Optional[SessionOnEnd] = None


@register
@dataclass
class SessionStatus(DocBase):
    state: str = field(default=STATE_PENDING, metadata={"omitempty": False})
    started_at: str = ""
    deadline: str = ""
    last_activity_at: str = ""
    ended_at: str = ""
    message: str = ""
    gpu_ids: List[int] = field(default_factory=list)


@register
@dataclass
class SessionDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_SESSION, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: SessionSpec = field(default_factory=SessionSpec)
    status: SessionStatus = field(default_factory=SessionStatus)


# ---------------------------------------------------------------------------
# Secret / Blueprint / Config / Volume / configurations
# ---------------------------------------------------------------------------
@register
@dataclass
class SecretSpec(DocBase):
    realm_id: str = ""
    space_id: str = ""
    data: Dict[str, str] = field(default_factory=dict,
                                 metadata={"omitempty": False})


@register
@dataclass
class SecretDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_SECRET, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: SecretSpec = field(default_factory=SecretSpec)


@register
@dataclass
class BlueprintParam(DocBase):
    name: str = ""
    default: str = ""
    required: bool = False
    description: str = ""


@register
@dataclass
class CellBlueprintSpec(DocBase):
    realm_id: str = ""
    space_id: str = ""
    params: List[BlueprintParam] = field(default_factory=list)
    name_prefix: str = ""
    template: Dict[str, Any] = field(default_factory=dict,
                                     metadata={"omitempty": False})


@register
@dataclass
class CellBlueprintDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_CELL_BLUEPRINT,
                      metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: CellBlueprintSpec = field(default_factory=CellBlueprintSpec)


@register
@dataclass
class CellConfigSpec(DocBase):
    realm_id: str = ""
    space_id: str = ""
    blueprint: str = field(default="", metadata={"omitempty": False})
    values: Dict[str, str] = field(default_factory=dict)
    env: List[str] = field(default_factory=list)
    name_prefix: str = ""


@register
@dataclass
class CellConfigDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_CELL_CONFIG, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: CellConfigSpec = field(default_factory=CellConfigSpec)


@register
@dataclass
class VolumeSpec(DocBase):
    realm_id: str = ""
    space_id: str = ""
    size_bytes: int = field(default=0, metadata={"omitzero": True})


@register
@dataclass
class VolumeStatus(DocBase):
    state: str = field(default=STATE_PENDING, metadata={"omitempty": False})
    path: str = ""


@register
@dataclass
class VolumeDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_VOLUME, metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: VolumeSpec = field(default_factory=VolumeSpec)
    status: VolumeStatus = field(default_factory=VolumeStatus)


@register
@dataclass
class ServerConfigurationSpec(DocBase):
    reconcile_interval_seconds: int = field(default=30,
                                            metadata={"omitempty": False})
    default_memory_limit_bytes: int = field(default=0,
                                            metadata={"omitzero": True})
    disk_pressure_warn_percent: int = field(default=85,
                                            metadata={"omitempty": False})
    disk_pressure_block_percent: int = field(default=95,
                                             metadata={"omitempty": False})
    tty_log_level: str = ""
    # MI355X: devices visible to the session scheduler ([] = autodetect)
    gpu_devices: List[int] = field(default_factory=list)


@register
@dataclass
class ServerConfigurationDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_SERVER_CONFIGURATION,
                      metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: ServerConfigurationSpec = field(
        default_factory=ServerConfigurationSpec)


@register
@dataclass
class ClientConfigurationSpec(DocBase):
    socket: str = ""
    default_realm: str = ""
    default_space: str = ""
    default_stack: str = ""


@register
@dataclass
class ClientConfigurationDoc(DocBase):
    api_version: str = field(default=API_VERSION, metadata={"omitempty": False})
    kind: str = field(default=KIND_CLIENT_CONFIGURATION,
                      metadata={"omitempty": False})
    metadata: Metadata = field(default_factory=Metadata)
    spec: ClientConfigurationSpec = field(
        default_factory=ClientConfigurationSpec)


DOC_TYPES = {
    KIND_REALM: RealmDoc,
    KIND_SPACE: SpaceDoc,
    KIND_STACK: StackDoc,
    KIND_CELL: CellDoc,
    KIND_CONTAINER: ContainerDoc,
    KIND_SESSION: SessionDoc,
    KIND_SECRET: SecretDoc,
    KIND_CELL_BLUEPRINT: CellBlueprintDoc,
    KIND_CELL_CONFIG: CellConfigDoc,
    KIND_VOLUME: VolumeDoc,
    KIND_SERVER_CONFIGURATION: ServerConfigurationDoc,
    KIND_CLIENT_CONFIGURATION: ClientConfigurationDoc,
}
