from .icl import evaluate_icl_tasks, gauntlet_composite, load_jsonl_task

__all__ = ["evaluate_icl_tasks", "gauntlet_composite", "load_jsonl_task"]
