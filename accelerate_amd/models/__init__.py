from .bert import BertConfig, BertForSequenceClassification

__all__ = ["BertConfig", "BertForSequenceClassification"]
