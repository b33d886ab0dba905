"""quoracle_amd — MI355X-native recursive consensus-agent orchestrator.

A from-scratch framework with the capabilities of shelvick/quoracle
(/root/reference): hierarchical agent trees where every decision is made by
multi-LLM consensus.  Instead of a Phoenix/OTP web app calling remote provider
APIs, the model pool is hosted locally on MI355X GPUs — one model (or TP
shard) per GPU — with hand-written CDNA4 HIP kernels for the hot ops and RCCL
over xGMI for cross-GPU traffic.

Layer map (see SURVEY.md §1 for the reference's equivalents):
  persistence/   SQLite-backed store            (ref: L0 Ecto/Postgres)
  engine/        local GPU inference engine     (ref: L1 ReqLLM HTTP layer)
  consensus/     decision pipeline              (ref: L2 lib/quoracle/consensus)
  agent/         asyncio actor runtime          (ref: L3 GenServer tree)
  actions/       capability execution           (ref: L4 lib/quoracle/actions)
  governance/    groves/profiles/skills/security(ref: L5)
  tasks/         task lifecycle + resume        (ref: L6)
  web/           monitor API                    (ref: L7 Phoenix LiveView)
  events.py      pub/sub backbone               (ref: LX Phoenix.PubSub)
  parallel/      RCCL/xGMI data plane + gloo control plane (new; ref has none)
  ops/           HIP/CDNA4 kernels (gfx950)     (new; ref has no GPU code)
"""

__version__ = "0.1.0"
