"""MI355X-native engine package."""
from fugue_amd.hip.registry import register_hip_engine

register_hip_engine()
