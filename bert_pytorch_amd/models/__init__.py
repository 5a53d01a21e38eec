from .bert import (  # noqa: F401
    BertConfig,
    BertEmbeddings,
    BertEncoder,
    BertForMaskedLM,
    BertForMultipleChoice,
    BertForNextSentencePrediction,
    BertForPreTraining,
    BertForQuestionAnswering,
    BertForSequenceClassification,
    BertForTokenClassification,
    BertLayer,
    BertModel,
    BertPooler,
    BertPreTrainedModel,
    LinearActivation,
)
from .criterion import BertPretrainingCriterion  # noqa: F401
