from .history import ContentBase, History
from .datasets import PairwisePreferenceDataset, PromptDataset, PromptTensorDictLoader
from .kl_controllers import AdaptiveKLController, ConstantKLController, KLControllerBase
from .rlhf_utils import (
    PromptData,
    PromptTensorDictTokenizer,
    RewardData,
    RolloutFromModel,
    TensorDictTokenizer,
    TokenizedDatasetLoader,
    TopKRewardSelector,
)
