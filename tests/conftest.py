import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


@pytest.fixture
def tiny_llm_config():
    return {
        "model": {
            "d_model": 64,
            "n_heads": 4,
            "n_layers": 2,
            "expansion_ratio": 4,
            "max_seq_len": 64,
            "vocab_size": 512,
            "attn_config": {"attn_impl": "torch"},
        },
        "optimizer": {
            "name": "adopt",
            "lr": 3e-4,
            "betas": [0.9, 0.9999],
            "eps": 1e-6,
            "weight_decay": 0.0,
        },
        "scheduler": {
            "schedulers": {
                "lr": {
                    "name": "cosine_with_warmup",
                    "t_warmup": "2ba",
                    "alpha_f": 0.1,
                    "t_max": "50ba",
                }
            }
        },
        "precision": "fp32",
        "device_train_microbatch_size": 2,
        "global_train_batch_size": 4,
        "algorithms": {
            "gradient_clipping": {"clipping_type": "norm", "clipping_threshold": 1.0}
        },
        "max_duration": "1000ba",
        "local_steps": "2ba",
        "seed": 17,
        "device_eval_batch_size": 2,
        "eval_subset_num_batches": 2,
        "max_seq_len": 64,
    }


@pytest.fixture
def tiny_cfg(tiny_llm_config, tmp_path):
    return {
        "run_uuid": "testrun",
        "seed": 1337,
        "use_wandb": False,
        "photon": {
            "saving_path": str(tmp_path / "ck"),
            "checkpoint": True,
            "resume_round": -1,
        },
        "fl": {
            "n_total_clients": 2,
            "n_clients_per_round": 2,
            "n_rounds": 2,
            "eval_period": 1,
            "strategy_name": "NESTOROV",
            "strategy_kwargs": {"server_learning_rate": 0.7, "server_momentum": 0.7},
            "reset_optimizer": True,
            "accept_failures_cnt": 0,
            "ignore_failed_rounds": False,
            "set_trainer_params_filter_keys": True,
            "set_trainer_key_to_filter": "transformer",
            "use_noise_scale_metric": False,
            "noise_scale_beta": 0.99,
        },
        "llm_config": tiny_llm_config,
        "dataset": {
            "train": {
                "streams": None,
                "split": "train",
                "root_local": "synthetic",
                "synthetic": True,
            },
            "val": {
                "streams": None,
                "split": "validation",
                "root_local": "synthetic",
                "synthetic": True,
            },
        },
        "wandb": {"setup": {}},
    }


@pytest.fixture(autouse=True)
def _deterministic():
    torch.manual_seed(0)


def free_port() -> int:
    """Pick an OS-assigned free TCP port (avoids fixed-port flakes when the
    suite runs back-to-back or alongside other jobs)."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
