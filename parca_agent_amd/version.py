__version__ = "0.2.0"

# Agent revision reported in the `profiler_agent_revision` metadata label
# (reference: reporter/metadata/agent.go:14-21).
REVISION = __version__
