from .bert_model import BertModel
