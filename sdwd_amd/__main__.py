"""``python -m sdwd_amd`` starts the API server on all visible GPUs."""
from .api.server import main

if __name__ == "__main__":
    main()
