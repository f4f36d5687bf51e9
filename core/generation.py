"""Repo-root entrypoint shim: `python -m core.generation --run R --prompt P`."""
from mlx_cuda_distributed_pretraining_amd.core.generation import main  # noqa: F401

if __name__ == "__main__":
    main()
