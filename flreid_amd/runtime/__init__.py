from flreid_amd.runtime.experiment import ExperimentStage
from flreid_amd.runtime.log import ExperimentLog
