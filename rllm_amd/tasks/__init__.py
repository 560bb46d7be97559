from rllm_amd.tasks.loader import load_dataset_config, load_tasks, resolve_verifier_ref

__all__ = ["load_tasks", "load_dataset_config", "resolve_verifier_ref"]
