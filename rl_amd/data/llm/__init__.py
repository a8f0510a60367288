from .history import ContentBase, History
from .datasets import PairwisePreferenceDataset, PromptDataset, PromptTensorDictLoader
