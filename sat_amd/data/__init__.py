from .dataset import DataSet, prepare_train_data, prepare_eval_data, prepare_test_data
from .vocabulary import Vocabulary
