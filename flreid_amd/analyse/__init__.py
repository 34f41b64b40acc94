from flreid_amd.analyse.accuracy import accuracy_on_round, accuracy_per_task
from flreid_amd.analyse.forgetting import forgetting_per_client, mean_forgetting
