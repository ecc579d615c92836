"""Typed service settings: YAML file < ``DETECTMATE_*`` environment overrides.

Capability parity with the reference ``ServiceSettings``
(/root/reference/src/service/settings.py:40-173):

* component identity (name/id/type/config_class) with a stable UUIDv5
  component id (settings.py:98-114),
* engine channel configuration (``engine_addr``, autostart, recv timeout,
  retry count, buffer size; settings.py:61-65),
* typed engine addresses rejecting unknown schemes (settings.py:30-37),
* ``out_addr`` fan-out list + dial timeout (settings.py:68-70),
* TLS input/output blocks with cross-validation that ``tls+tcp`` addresses
  have a matching TLS block (settings.py:116-132),
* HTTP admin host/port (settings.py:77-78),
* ``from_yaml`` merging YAML below env vars (settings.py:134-168), env
  prefix ``DETECTMATE_`` with ``__`` as the nested delimiter and unknown
  keys rejected (settings.py:80-84).

MI355X-first additions (this framework batches the engine loop and runs
stage compute on GPU — SURVEY.md §7):

* ``engine_batch_size`` / ``engine_batch_linger_ms``: the engine drains up
  to ``engine_batch_size`` frames (waiting at most the linger) and hands the
  whole batch to the component — the single biggest throughput lever over
  the reference's strictly per-message loop (engine.py:196-264).
* ``device``: torch device for stage compute (``cuda:N`` or ``cpu``);
  ``None`` auto-selects cuda when available.
* ``dist_*``: process-group knobs used when a pipeline is placed across
  GPUs with RCCL over xGMI instead of socket hops.

Implemented on plain pydantic (pydantic-settings is not available in this
environment); the env-merge logic is explicit in :meth:`ServiceSettings.from_yaml`.
"""
from __future__ import annotations

import os
import re
import uuid
from pathlib import Path
from typing import Any, List, Optional

import yaml
from pydantic import BaseModel, ConfigDict, Field, field_validator, model_validator

ENV_PREFIX = "DETECTMATE_"
ENV_NESTED_DELIMITER = "__"

#: Accepted engine address schemes (reference settings.py:30-37).
#: Wire framings: tcp/tls+tcp speak the NNG SP mapping, ws speaks
#: RFC 6455, ipc/inproc use the compact intra-node framing — see
#: engine/sockets.py.
_ADDR_SCHEMES = ("ipc", "tcp", "tls+tcp", "ws", "inproc", "shm")
_ADDR_RE = re.compile(r"^(?P<scheme>[a-z+]+)://(?P<rest>.+)$")


class EngineAddr(str):
    """A validated engine address: ``scheme://rest`` with a known scheme."""

    @classmethod
    def validate(cls, value: str) -> "EngineAddr":
        if not isinstance(value, str):
            raise ValueError(f"address must be a string, got {type(value)!r}")
        m = _ADDR_RE.match(value)
        if not m:
            raise ValueError(f"invalid engine address {value!r} (expected scheme://...)")
        scheme = m.group("scheme")
        if scheme not in _ADDR_SCHEMES:
            raise ValueError(
                f"unsupported scheme {scheme!r} in {value!r}; supported: {_ADDR_SCHEMES}"
            )
        if scheme in ("tcp", "tls+tcp", "ws"):
            host_port = m.group("rest")
            if ":" not in host_port.rsplit("]", 1)[-1]:
                raise ValueError(f"{scheme} address {value!r} must include a port")
        return cls(value)

    @property
    def scheme(self) -> str:
        return _ADDR_RE.match(self).group("scheme")  # type: ignore[union-attr]

    @property
    def rest(self) -> str:
        return _ADDR_RE.match(self).group("rest")  # type: ignore[union-attr]


def _validate_addr(value: Any) -> EngineAddr:
    return EngineAddr.validate(value)


class TlsInputConfig(BaseModel):
    """TLS config for the listening (input) socket.

    Reference parity: settings.py:11-18 (``cert_key_file`` served by the
    listener). Implemented with the stdlib ``ssl`` module.
    """

    model_config = ConfigDict(extra="forbid")
    cert_key_file: Path


class TlsOutputConfig(BaseModel):
    """TLS config for dialing (output) sockets.

    Reference parity: settings.py:20-28 (``ca_file`` + ``server_name`` SNI).
    """

    model_config = ConfigDict(extra="forbid")
    ca_file: Path
    server_name: str = "localhost"


class ServiceSettings(BaseModel):
    model_config = ConfigDict(extra="forbid", validate_assignment=True)

    # --- component identity (reference settings.py:49-52) ---
    component_name: Optional[str] = None
    component_id: Optional[str] = None
    component_type: str = "core"
    config_class: Optional[str] = None

    # --- logging (reference settings.py:55-58) ---
    log_level: str = "INFO"
    log_dir: Path = Path("logs")

    # --- engine channel (reference settings.py:61-65) ---
    engine_addr: str = "ipc:///tmp/detectmate.engine.ipc"
    engine_autostart: bool = True
    engine_recv_timeout: int = Field(default=100, ge=1, description="recv poll timeout, ms")
    engine_retry_count: int = Field(default=10, ge=1)
    engine_buffer_size: int = Field(default=128, ge=0, le=8192, description="per-socket queued frames")

    #: source mode: the component GENERATES frames (readers tailing a
    #: file) instead of receiving them on the engine socket
    engine_source_mode: bool = False

    # --- batching (MI355X-native; no reference equivalent) ---
    engine_batch_size: int = Field(default=256, ge=1, le=1_048_576)
    engine_batch_linger_ms: float = Field(default=2.0, ge=0.0)

    #: packed data plane: socket reader threads decode LogSchema frames
    #: straight into tensors in C++ (zero Python objects per frame).
    #: Requires a plain ipc/tcp listener and a component exposing
    #: ``process_packed_frames`` (e.g. FusedPipelineDetector); the engine
    #: falls back to the frame loop otherwise.
    engine_packed_mode: bool = False

    # --- outputs (reference settings.py:68-70) ---
    out_addr: List[str] = Field(default_factory=list)
    dial_timeout: int = Field(default=1000, ge=0, description="output dial timeout, ms")

    # --- TLS (reference settings.py:73-74) ---
    tls_input: Optional[TlsInputConfig] = None
    tls_output: Optional[TlsOutputConfig] = None

    # --- admin HTTP (reference settings.py:77-78) ---
    http_host: str = "127.0.0.1"
    http_port: int = Field(default=8000, ge=0, le=65535)
    http_enabled: bool = True

    # --- component config file (reference settings.py:53) ---
    config_file: Optional[Path] = None

    #: directory that /admin/checkpoint + /admin/restore may read/write.
    #: Unset => the admin endpoints refuse (the Python API is unaffected):
    #: the admin HTTP surface is unauthenticated, so letting callers name
    #: arbitrary filesystem paths would allow checkpoint planting/probing.
    checkpoint_dir: Optional[Path] = None

    # --- MI355X-native compute placement ---
    device: Optional[str] = None
    dist_backend: Optional[str] = None
    dist_world_size: int = Field(default=1, ge=1)
    dist_rank: int = Field(default=0, ge=0)

    #: settings-driven distributed placement (reference principle:
    #: topology is config — container/config/parser_settings.yaml wires
    #: stages by address; here `torchrun ... detectmate --settings x.yaml`
    #: wires them by rank):
    #:   "dp"     N independent services; NewValue state merges on
    #:            dp_sync (admin POST /admin/dp-sync or collective cadence)
    #:   "fanout" rank dist_src_rank ingests + processes, broadcasts its
    #:            outputs to every other rank (the reference multi_output
    #:            1->N as ONE collective over xGMI)
    #:   "stage"  chain: rank 0 ingests from its socket, each rank
    #:            processes and forwards to rank+1, the last rank emits
    #:            to its out_addr (pipeline placement, P2P hops)
    #: Addresses and component_name may contain "{rank}"/"{world}"
    #: placeholders, substituted after the process group initializes.
    dist_mode: Optional[str] = None
    dist_src_rank: int = Field(default=0, ge=0)
    #: collective timeout: a dead peer surfaces as an error after this
    #: long, and the engine's dist loops DEGRADE to the socket path
    #: (drop-don't-block posture on RCCL — SURVEY.md §5.8 hard part)
    dist_timeout_s: float = Field(default=120.0, gt=0)

    @field_validator("dist_mode")
    @classmethod
    def _check_dist_mode(cls, v):
        if v is not None and v not in ("dp", "fanout", "stage"):
            raise ValueError(f"dist_mode must be dp|fanout|stage, got {v!r}")
        return v

    def resolve_dist_placeholders(self, rank: int, world: int) -> "ServiceSettings":
        """Substitute {rank}/{world} in addresses and identity; the
        component_id regenerates from the substituted identity so each
        rank gets a stable, distinct id."""
        d = self.model_dump()
        def sub(v):
            return (v.replace("{rank}", str(rank)).replace("{world}", str(world))
                    if isinstance(v, str) else v)
        changed = False
        for key in ("engine_addr", "component_name"):
            nv = sub(d.get(key))
            if nv != d.get(key):
                d[key] = nv
                changed = True
        new_out = [sub(a) for a in d.get("out_addr", [])]
        if new_out != d.get("out_addr"):
            d["out_addr"] = new_out
            changed = True
        d["dist_rank"], d["dist_world_size"] = rank, world
        if changed:
            d["component_id"] = None  # regenerate per-rank identity
        return ServiceSettings.model_validate(d)

    # ------------------------------------------------------------------
    @field_validator("engine_addr")
    @classmethod
    def _check_engine_addr(cls, v: str) -> str:
        return str(_validate_addr(v))

    @field_validator("out_addr")
    @classmethod
    def _check_out_addrs(cls, v: List[str]) -> List[str]:
        return [str(_validate_addr(a)) for a in v]

    @model_validator(mode="after")
    def _ensure_component_id(self) -> "ServiceSettings":
        """Stable component identity (reference settings.py:98-114).

        UUIDv5 of ``detectmate/{type}/{name}`` when a name is set, else of
        ``detectmate/{type}|{engine_addr}``.
        """
        if self.component_id is None:
            if self.component_name:
                seed = f"detectmate/{self.component_type}/{self.component_name}"
            else:
                seed = f"detectmate/{self.component_type}|{self.engine_addr}"
            object.__setattr__(
                self, "component_id", str(uuid.uuid5(uuid.NAMESPACE_DNS, seed))
            )
        return self

    @model_validator(mode="after")
    def _validate_tls_config_present(self) -> "ServiceSettings":
        """``tls+tcp`` addresses require the matching TLS block (settings.py:116-132)."""
        if EngineAddr(self.engine_addr).scheme == "tls+tcp" and self.tls_input is None:
            raise ValueError(
                "engine_addr uses tls+tcp but no tls_input configuration is present"
            )
        for addr in self.out_addr:
            if EngineAddr(addr).scheme == "tls+tcp" and self.tls_output is None:
                raise ValueError(
                    f"out_addr {addr!r} uses tls+tcp but no tls_output configuration is present"
                )
        return self

    # ------------------------------------------------------------------
    @classmethod
    def _env_overrides(cls) -> dict:
        """Collect ``DETECTMATE_*`` env vars into a nested dict.

        ``DETECTMATE_ENGINE_ADDR=tcp://...`` → ``{"engine_addr": ...}``;
        ``DETECTMATE_TLS_INPUT__CERT_KEY_FILE=...`` →
        ``{"tls_input": {"cert_key_file": ...}}`` (reference settings.py:80-84).
        """
        out: dict = {}
        list_fields = {"out_addr"}
        for key, raw in os.environ.items():
            if not key.startswith(ENV_PREFIX):
                continue
            path = key[len(ENV_PREFIX):].lower().split(ENV_NESTED_DELIMITER)
            if path[0] not in cls.model_fields:
                continue
            value: Any = raw
            if path[0] in list_fields:
                try:
                    value = yaml.safe_load(raw)
                except yaml.YAMLError:
                    value = raw
                if isinstance(value, str):
                    value = [a.strip() for a in value.split(",") if a.strip()]
            else:
                # YAML-parse scalars so "true"/"100" become bool/int.
                try:
                    value = yaml.safe_load(raw)
                except yaml.YAMLError:
                    value = raw
            node = out
            for part in path[:-1]:
                node = node.setdefault(part, {})
            node[path[-1]] = value
        return out

    @classmethod
    def from_yaml(cls, path: str | Path) -> "ServiceSettings":
        """Load settings from a YAML file with env-var precedence.

        Reference parity: settings.py:134-168 (env vars win over YAML keys).
        """
        path = Path(path)
        with open(path, "r", encoding="utf-8") as fh:
            data = yaml.safe_load(fh) or {}
        if not isinstance(data, dict):
            raise ValueError(f"settings file {path} must contain a YAML mapping")
        env = cls._env_overrides()
        merged = _deep_merge(data, env)
        return cls.model_validate(merged)

    @classmethod
    def from_env(cls) -> "ServiceSettings":
        return cls.model_validate(cls._env_overrides())


def _deep_merge(base: dict, override: dict) -> dict:
    out = dict(base)
    for k, v in override.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = v
    return out
