"""Synthetic audit-log workload generator.

There is no network access for datasets, so benches and integration tests
generate Linux-audit-shaped log lines (same structure as the reference's
tests/library_integration/audit.log fixture: a ``type=... msg=audit(ts:id):``
header followed by key=value content) with a small template family and
controllable anomaly rate. Written fresh for this repo — lines are
generated, not copied.
"""
from __future__ import annotations

import random
from typing import List, Tuple

#: Content templates with ``<*>`` wildcards — these drive both generation
#: (wildcards filled from value pools) and the MatcherParser template set.
AUDIT_TEMPLATES: List[str] = [
    "pid=<*> uid=<*> auid=<*> ses=<*> msg='op=<*> acct=<*> exe=<*> hostname=<*> addr=<*> terminal=<*> res=<*>'",
    "pid=<*> uid=<*> auid=<*> ses=<*> msg='unit=<*> comm=<*> exe=<*> hostname=<*> addr=<*> terminal=<*> res=<*>'",
    "pid=<*> uid=<*> old-auid=<*> auid=<*> tty=<*> old-ses=<*> ses=<*> res=<*>",
    "arch=<*> syscall=<*> success=<*> exit=<*> a0=<*> a1=<*> items=<*> ppid=<*> pid=<*> auid=<*> uid=<*> gid=<*> comm=<*> exe=<*> key=<*>",
    "proctitle=<*>",
    "table=<*> family=<*> entries=<*>",
    "pid=<*> uid=<*> auid=<*> ses=<*> msg='cwd=<*> cmd=<*> terminal=<*> res=<*>'",
    "audit_pid=<*> old=<*> auid=<*> ses=<*> res=<*>",
]

#: ``log_format`` for the audit header; content follows the final colon-space.
AUDIT_LOG_FORMAT = "type=<Type> msg=audit(<Time>:<Serial>): <Content>"

_TYPES = [
    "USER_ACCT", "CRED_ACQ", "LOGIN", "USER_START", "CRED_DISP",
    "USER_END", "SYSCALL", "PROCTITLE", "SERVICE_START", "NETFILTER_CFG",
]
_EXES = ["/usr/sbin/cron", "/usr/sbin/sshd", "/usr/bin/sudo", "/bin/login"]
_ACCTS = ["root", "daemon", "www-data", "backup", "operator"]
_OPS = ["PAM:accounting", "PAM:setcred", "PAM:session_open", "PAM:session_close"]
_COMMS = ["cron", "sshd", "systemd", "bash", "apt-get"]


class AuditLogGenerator:
    """Deterministic (seeded) generator of audit-shaped log lines."""

    def __init__(self, seed: int = 1234, anomaly_rate: float = 0.0) -> None:
        self.rng = random.Random(seed)
        self.anomaly_rate = anomaly_rate
        self._serial = 100

    def _fill(self, template: str, anomalous: bool) -> str:
        rng = self.rng
        out = []
        parts = template.split("<*>")
        for i, lit in enumerate(parts):
            out.append(lit)
            if i == len(parts) - 1:
                break
            prev = lit.rstrip()
            if prev.endswith("exe="):
                v = "/usr/bin/evil" if anomalous else rng.choice(_EXES)
            elif prev.endswith("acct="):
                v = f'"{ "intruder" if anomalous else rng.choice(_ACCTS)}"'
            elif prev.endswith("op="):
                v = rng.choice(_OPS)
            elif prev.endswith("comm="):
                v = rng.choice(_COMMS)
            elif prev.endswith("res="):
                v = "failed" if anomalous else "success"
            elif prev.endswith(("hostname=", "addr=", "terminal=")):
                v = rng.choice(["?", "cron", "pts/0"])
            else:
                v = str(rng.randint(0, 65535))
            out.append(v)
        return "".join(out)

    def line(self) -> Tuple[str, bool, int]:
        """Returns (line, is_anomalous, template_index)."""
        rng = self.rng
        tidx = rng.randrange(len(AUDIT_TEMPLATES))
        anomalous = rng.random() < self.anomaly_rate
        content = self._fill(AUDIT_TEMPLATES[tidx], anomalous)
        ts = 1642723741 + self._serial // 7
        line = f"type={rng.choice(_TYPES)} msg=audit({ts}.{self._serial % 1000:03d}:{self._serial}): {content}"
        self._serial += 1
        return line, anomalous, tidx

    def lines(self, n: int) -> List[str]:
        return [self.line()[0] for _ in range(n)]
