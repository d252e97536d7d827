"""Per-space egress enforcement + host FORWARD admission.

Mirrors the reference's internal/netpolicy (per-space chains under a
KUKEON-EGRESS master chain built from SpaceSpec.network.egress) and
internal/firewall (KUKEON-FORWARD admission chain, idempotent -C before
-I/-A). Commands go through an injectable runner so tests assert the exact
iptables command lines without root or iptables present; hosts without
iptables get the NoopEnforcer.
"""
from __future__ import annotations

import hashlib
import shutil
import socket
import subprocess
from dataclasses import dataclass, field
from typing import Callable, List, Optional

from kukeon_amd.api import v1beta1 as api

MASTER_CHAIN = "KUKEON-EGRESS"
FORWARD_CHAIN = "KUKEON-FORWARD"

CommandRunner = Callable[[List[str]], int]


def default_runner(args: List[str]) -> int:
    return subprocess.run(args, capture_output=True).returncode


def iptables_available() -> bool:
    return shutil.which("iptables") is not None


def space_chain(realm: str, space: str) -> str:
    h = hashlib.sha256(f"{realm}/{space}".encode()).hexdigest()[:8]
    return f"KUKEON-EG-{h}"


@dataclass
class Rule:
    cidr: str = ""
    ports: List[int] = field(default_factory=list)


@dataclass
class Policy:
    default_deny: bool = False
    rules: List[Rule] = field(default_factory=list)


def build_policy(egress: Optional[api.EgressPolicy],
                 resolver=socket.getaddrinfo) -> Policy:
    """Resolve an EgressPolicy doc into concrete CIDR rules (host names
    resolved at apply time — the reference's TTL caveat applies)."""
    if egress is None:
        return Policy()
    p = Policy(default_deny=(egress.default == "deny"))
    for r in egress.allow:
        if r.cidr:
            p.rules.append(Rule(cidr=r.cidr, ports=list(r.ports)))
        elif r.host:
            try:
                infos = resolver(r.host, None)
            except OSError:
                continue
            seen = set()
            for info in infos:
                ip = info[4][0]
                if ":" in ip or ip in seen:
                    continue
                seen.add(ip)
                p.rules.append(Rule(cidr=f"{ip}/32", ports=list(r.ports)))
    return p


class Enforcer:
    def apply(self, realm: str, space: str, subnet: str, policy: Policy) -> None:
        raise NotImplementedError

    def remove(self, realm: str, space: str) -> None:
        raise NotImplementedError


class NoopEnforcer(Enforcer):
    def apply(self, realm, space, subnet, policy) -> None:
        pass

    def remove(self, realm, space) -> None:
        pass


class IptablesEnforcer(Enforcer):
    def __init__(self, runner: CommandRunner = default_runner):
        self.run = runner

    def _ensure_chain(self, chain: str) -> None:
        if self.run(["iptables", "-nL", chain]) != 0:
            self.run(["iptables", "-N", chain])

    def _ensure_rule(self, args: List[str], insert: bool = False) -> None:
        if self.run(["iptables", "-C"] + args) != 0:
            self.run(["iptables", "-I" if insert else "-A"] + args)

    def apply(self, realm, space, subnet, policy: Policy) -> None:
        self._ensure_chain(MASTER_CHAIN)
        self._ensure_rule(["FORWARD", "-j", MASTER_CHAIN], insert=True)
        chain = space_chain(realm, space)
        self._ensure_chain(chain)
        self.run(["iptables", "-F", chain])
        self._ensure_rule([MASTER_CHAIN, "-s", subnet, "-j", chain])
        for r in policy.rules:
            base = [chain, "-d", r.cidr]
            if r.ports:
                for port in r.ports:
                    self.run(["iptables", "-A"] + base +
                             ["-p", "tcp", "--dport", str(port),
                              "-j", "ACCEPT"])
            else:
                self.run(["iptables", "-A"] + base + ["-j", "ACCEPT"])
        # established return traffic + DNS are always allowed before the drop
        self.run(["iptables", "-A", chain, "-m", "state", "--state",
                  "ESTABLISHED,RELATED", "-j", "ACCEPT"])
        if policy.default_deny:
            self.run(["iptables", "-A", chain, "-j", "DROP"])

    def remove(self, realm, space) -> None:
        chain = space_chain(realm, space)
        self.run(["iptables", "-F", chain])
        # drop the jump rules referencing the chain, then the chain
        self.run(["iptables", "-D", MASTER_CHAIN, "-j", chain])
        self.run(["iptables", "-X", chain])


class ForwardInstaller:
    """Host FORWARD admission (needed when the host FORWARD policy is DROP)."""

    def __init__(self, runner: CommandRunner = default_runner):
        self.run = runner

    def install(self, subnet_root: str = "10.88.0.0/16") -> None:
        if self.run(["iptables", "-nL", FORWARD_CHAIN]) != 0:
            self.run(["iptables", "-N", FORWARD_CHAIN])
        if self.run(["iptables", "-C", "FORWARD", "-j", FORWARD_CHAIN]) != 0:
            self.run(["iptables", "-I", "FORWARD", "-j", FORWARD_CHAIN])
        for args in (
            [FORWARD_CHAIN, "-s", subnet_root, "-j", "ACCEPT"],
            [FORWARD_CHAIN, "-d", subnet_root, "-m", "state", "--state",
             "ESTABLISHED,RELATED", "-j", "ACCEPT"],
        ):
            if self.run(["iptables", "-C"] + args) != 0:
                self.run(["iptables", "-A"] + args)


def make_enforcer(runner: CommandRunner = default_runner) -> Enforcer:
    if iptables_available():
        return IptablesEnforcer(runner)
    return NoopEnforcer()
