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
from .tf_import import load_tf_weights, tf_name_to_state_key  # noqa: F401
