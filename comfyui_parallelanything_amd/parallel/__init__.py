from .chain import DeviceChain, available_devices, chain_append, chain_from_slots, normalize_weights  # noqa: F401
from .engine import ParallelEngine, WorkerError, install_parallel_forward, uninstall_parallel_forward  # noqa: F401
from .cleanup import cleanup_parallel_model, register_finalizer, aggressive_cleanup  # noqa: F401
from .pipeline import configure_pipeline, set_pipeline_mode, pipeline_mode_active  # noqa: F401
from .replicate import replicate_module, broadcast_module  # noqa: F401
