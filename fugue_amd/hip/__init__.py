"""MI355X-native engine package."""
from fugue_amd.hip.registry import register_hip_engine
from fugue_amd.hip.udf import register_device_params

register_hip_engine()
register_device_params()
