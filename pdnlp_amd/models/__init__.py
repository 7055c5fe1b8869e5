from .bert import (  # noqa: F401
    BertForSequenceClassification,
    RobertaForSequenceClassification,
    SequenceClassifierOutput,
    build_model,
)
