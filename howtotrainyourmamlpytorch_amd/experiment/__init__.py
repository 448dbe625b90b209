from .storage import (  # noqa: F401
    build_experiment_folder,
    load_from_json,
    load_statistics,
    save_statistics,
    save_to_json,
)
