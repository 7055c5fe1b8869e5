from .dataset import load_data, train_dev_split, ClsDataset, SyntheticClsDataset, LABELS, label2id, id2label  # noqa: F401
from .tokenizer import BertWordPieceTokenizer, CharTokenizer, build_tokenizer  # noqa: F401
from .collate import Collate  # noqa: F401
from .sampler import DistributedSampler  # noqa: F401
