from .episodes import FewShotEpisodeDataset, scan_class_folders, split_classes  # noqa: F401
from .loader import MetaLearningSystemDataLoader  # noqa: F401
from .synthetic import SyntheticEpisodeStream  # noqa: F401
from .tools import maybe_unzip_dataset  # noqa: F401
