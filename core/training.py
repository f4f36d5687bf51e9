"""Repo-root entrypoint shim: `python -m core.training --config X`."""
from mlx_cuda_distributed_pretraining_amd.core.training import main, train  # noqa: F401

if __name__ == "__main__":
    main()
