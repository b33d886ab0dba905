"""TaskRuntime: the dependency bundle threaded through every component.

The reference passes registry / dynsup / pubsub / sandbox_owner explicitly
into every process for test isolation; this container is the
same idea — one per orchestrator instance, never global.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from ..engine.pool import EnginePool
from ..events import EventBus
from ..governance.profiles import ProfileStore
from ..governance.security import SecretVault
from ..governance.skills import SkillLoader
from ..persistence.store import Store
from ..registry import Registry


@dataclass
class RuntimeConfig:
    # Role-based model settings (reference: models/config_model_settings.ex):
    # which hosted model serves each auxiliary role.  Empty entries fall
    # back per call site (embedding -> pool embedder, answer_engine ->
    # first pool model, summarization -> the history-owning model itself).
    model_roles: Dict[str, str] = field(default_factory=dict)
    groves_dir: Optional[str] = None
    skills_dir: Optional[str] = None
    default_working_dir: str = "/tmp"
    shell_sync_threshold_s: float = 0.1   # smart-mode boundary (ref: shell.ex:13)
    action_timeout_s: float = 300.0
    shell_timeout_s: float = 600.0
    spawn_retries: int = 3
    consensus_retries: int = 3            # ref: message_handler.ex:353-421
    # hard cap on one model generate (a wedged engine degrades to a
    # per-model failure -> partial-pool consensus instead of a hung agent)
    generate_timeout_s: float = 600.0
    test_mode: bool = False
    # verbose prompt tracing: broadcast every sent message list + raw
    # response on agents:<id>:trace (reference: consensus_handler.ex:154-179
    # debug broadcasts + the show_llm_prompts tooling)
    trace_prompts: bool = False


class TaskRuntime:
    def __init__(
        self,
        *,
        store: Optional[Store] = None,
        bus: Optional[EventBus] = None,
        registry: Optional[Registry] = None,
        engines: Optional[EnginePool] = None,
        profiles: Optional[ProfileStore] = None,
        vault: Optional[SecretVault] = None,
        skills: Optional[SkillLoader] = None,
        config: Optional[RuntimeConfig] = None,
    ):
        self.store = store or Store(":memory:")
        self.bus = bus or EventBus()
        self.registry = registry or Registry()
        self.engines = engines or EnginePool()
        self.profiles = profiles or ProfileStore(self.store)
        self._vault = vault          # lazy: key material only needed on use
        self.config = config or RuntimeConfig()
        self.skills = skills or SkillLoader(self.config.skills_dir)
        self.supervisor: Any = None   # set by agent.supervisor.Supervisor
        self.extras: Dict[str, Any] = {}   # injectable test hooks (http_fn, ...)

    @property
    def vault(self) -> SecretVault:
        """Constructed on first secret use: SecretVault refuses to exist
        without key material (QUORACLE_VAULT_KEY or an injected vault), so
        secret-free runs never need a key."""
        if self._vault is None:
            self._vault = SecretVault(self.store)
        return self._vault

    @vault.setter
    def vault(self, value: Optional[SecretVault]) -> None:
        self._vault = value

    def embed_many(self, texts):
        return self.engines.embed_many_sync(texts)
