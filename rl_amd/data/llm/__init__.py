from .history import ContentBase, History
from .datasets import PairwisePreferenceDataset, PromptDataset, PromptTensorDictLoader
from .kl_controllers import AdaptiveKLController, ConstantKLController, KLControllerBase
