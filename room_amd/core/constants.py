"""Domain constants preserved from the reference (src/shared/constants.ts).

Pacing constants define the *configuration* under which agent cycles are
measured: throughput runs set gaps to MIN_CYCLE_GAP_MS.
"""

APP_NAME = "RoomAMD"
APP_ID = "ai.roomamd.room"

TRIGGER_TYPES = ("cron", "once", "manual", "webhook")
TASK_STATUSES = ("active", "paused", "completed")
ROOM_STATUSES = ("active", "paused", "stopped")
AGENT_STATES = ("idle", "thinking", "acting", "voting", "rate_limited", "blocked")
DECISION_TYPES = ("strategy", "resource", "personnel", "rule_change", "low_impact")
GOAL_STATUSES = ("active", "in_progress", "completed", "abandoned", "blocked")
WALLET_TX_TYPES = ("send", "receive", "fund", "purchase")

# Default pacing (reference: schema.ts:53, constants.ts:161-175)
DEFAULT_QUEEN_CYCLE_GAP_MS = 1_800_000  # 30 min
DEFAULT_QUEEN_MAX_TURNS = 50
MIN_CYCLE_GAP_MS = 1_000  # floor; throughput benches run at this
MOMENTUM_GAP_MS = 10_000  # adaptive gap when WIP exists (agent-loop.ts:204-217)

# Session continuity (agent-loop.ts:462-532)
SESSION_ROTATE_CYCLES = 20      # CLI-style session rotation
SESSION_COMPRESS_AT_MSGS = 30   # API-session compression threshold
SESSION_TRIM_TO_MSGS = 40       # hard trim

# Skills budget per cycle (skills.ts:5-6)
SKILLS_MAX_PER_CYCLE = 8
SKILLS_CHAR_BUDGET = 6000

# Memory injection (agent-loop.ts:585-602)
MEMORY_TOP_K = 5

# Rate limiting (rate-limit.ts:9-12)
RATE_LIMIT_MIN_WAIT_MS = 30_000
RATE_LIMIT_MAX_WAIT_MS = 3_600_000
RATE_LIMIT_MAX_RETRIES = 3

# Quorum (quorum.ts:35)
ANNOUNCE_DEFAULT_DELAY_MINUTES = 10

# Self-modification (self-mod.ts:5-34)
SELF_MOD_RATE_LIMIT_MS = 60_000

# Task runner (task-runner.ts:33,53-93)
TASK_SESSION_ROTATE_RUNS = 20
TASK_MIN_CONCURRENT = 1
TASK_MAX_CONCURRENT = 10

# Agent cycle timeouts (agent-loop.ts:748)
CYCLE_TIMEOUT_MS = 15 * 60 * 1000
CYCLE_TIMEOUT_EXECUTOR_MS = 30 * 60 * 1000

# Webhooks (webhooks.ts:16-17)
WEBHOOK_RATE_LIMIT_PER_MIN = 30

# Server rate limits (server/index.ts:384-386)
API_RATE_LIMIT_READ_PER_MIN = 300
API_RATE_LIMIT_WRITE_PER_MIN = 120

# Embeddings (embeddings.ts:33-34)
EMBEDDING_MODEL = "all-MiniLM-L6-v2"
EMBEDDING_DIM = 384

# Hybrid search fusion weights (db-queries.ts:1021-1059)
HYBRID_FTS_WEIGHT = 0.4
HYBRID_SEMANTIC_WEIGHT = 0.6
HYBRID_RRF_K = 60

# The pinned local model (local-model.ts:3-5 pins qwen3-coder:30b via Ollama;
# here it runs in-process on the GPU).
LOCAL_MODEL_TAG = "qwen3-coder-30b"

QUEEN_DEFAULTS_BY_PLAN = {
    "none": {"queenCycleGapMs": 10 * 60 * 1000, "queenMaxTurns": 50},
    "pro": {"queenCycleGapMs": 5 * 60 * 1000, "queenMaxTurns": 50},
    "max": {"queenCycleGapMs": 30 * 1000, "queenMaxTurns": 50},
    "api": {"queenCycleGapMs": 2 * 60 * 1000, "queenMaxTurns": 50},
}

# constants.ts:168-174 — codex/ChatGPT plan pacing map
CHATGPT_DEFAULTS_BY_PLAN = {
    "none": {"queenCycleGapMs": 10 * 60 * 1000, "queenMaxTurns": 50},
    "plus": {"queenCycleGapMs": 5 * 60 * 1000, "queenMaxTurns": 50},
    "pro": {"queenCycleGapMs": 2 * 60 * 1000, "queenMaxTurns": 50},
    "api": {"queenCycleGapMs": 2 * 60 * 1000, "queenMaxTurns": 50},
}

WORKER_ROLE_PRESETS = {
    "guardian": {
        "cycleGapMs": 30_000,
        "maxTurns": 30,
        "systemPromptPrefix": (
            "Monitor and observe. Focus on detecting anomalies. "
            "Do not spawn workers or make purchases."
        ),
    },
    "analyst": {
        "cycleGapMs": 60_000,
        "maxTurns": 100,
        "systemPromptPrefix": (
            "Perform deep analysis. Work to COMPLETION — you have plenty of turns.\n"
            "Save progress with room_save_wip before your cycle ends."
        ),
    },
    "writer": {
        "cycleGapMs": 60_000,
        "maxTurns": 100,
        "systemPromptPrefix": (
            "Produce high-quality written output. Work to COMPLETION — you have "
            "plenty of turns.\nSave progress with room_save_wip before your cycle ends."
        ),
    },
    "executor": {
        "cycleGapMs": 15_000,
        "maxTurns": 200,
        "systemPromptPrefix": (
            "You are an execution agent. Your ONLY job is to DO things — not plan, "
            "not coordinate.\n\nContinue from your WIP if you have one. Otherwise "
            "start your assigned tasks immediately.\nRun your full action chain to "
            "completion. You have plenty of turns — don't rush.\nSave progress with "
            "room_save_wip before your cycle ends.\nStore ALL results with "
            "room_remember so teammates can access them."
        ),
    },
    "researcher": {
        "cycleGapMs": 30_000,
        "maxTurns": 100,
        "systemPromptPrefix": (
            "You are a research specialist. Be data-driven: real numbers, URLs, "
            "pricing data.\nCheck room_recall before starting any topic — don't "
            "duplicate existing research.\nWork to COMPLETION. Message key findings "
            "to the keeper.\nSave progress with room_save_wip before your cycle ends."
        ),
    },
}

DEFAULT_ROOM_CONFIG = {
    "threshold": "majority",
    "timeoutMinutes": 60,
    "tieBreaker": "queen",
    "autoApprove": ["low_impact"],
    "minCycleGapMs": MIN_CYCLE_GAP_MS,
    "minVoters": 0,
    "sealedBallot": False,
    "voterHealth": False,
    "voterHealthThreshold": 0.5,
}

SUPPORTED_CHAINS = ("base", "ethereum", "arbitrum", "optimism", "polygon")
SUPPORTED_TOKENS = ("usdc", "usdt")

CHAIN_CONFIGS = {
    "base": {"chainId": 8453, "name": "Base", "rpcUrl": "https://mainnet.base.org"},
    "ethereum": {"chainId": 1, "name": "Ethereum", "rpcUrl": "https://eth.llamarpc.com"},
    "arbitrum": {"chainId": 42161, "name": "Arbitrum", "rpcUrl": "https://arb1.arbitrum.io/rpc"},
    "optimism": {"chainId": 10, "name": "Optimism", "rpcUrl": "https://mainnet.optimism.io"},
    "polygon": {"chainId": 137, "name": "Polygon", "rpcUrl": "https://polygon-rpc.com"},
    "base-sepolia": {"chainId": 84532, "name": "Base Sepolia", "rpcUrl": "https://sepolia.base.org"},
}
