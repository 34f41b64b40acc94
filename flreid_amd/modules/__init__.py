from flreid_amd.modules.client import ClientModule
from flreid_amd.modules.criterion import CriterionModule
from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.operator import OperatorModule
from flreid_amd.modules.server import ServerModule
