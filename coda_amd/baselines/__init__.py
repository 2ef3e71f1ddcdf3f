from .iid import IID
from .uncertainty import Uncertainty
from .activetesting import ActiveTesting
from .vma import VMA
from .modelpicker import ModelPicker, TASK_EPS

__all__ = ["IID", "Uncertainty", "ActiveTesting", "VMA", "ModelPicker",
           "TASK_EPS"]
