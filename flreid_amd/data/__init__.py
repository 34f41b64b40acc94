from flreid_amd.data.augment import augmentations
from flreid_amd.data.loader import ReIDImageDataset
from flreid_amd.data.pipeline import ReIDTaskPipeline
from flreid_amd.data.synthetic import SyntheticReIDDataset, materialize_task_dir
