from .trainer import (build_loader_model_grapher, execute_graph, train,
                      test, run)
from .grapher import Grapher
from .saver import CheckpointBundle, ModelSaver, get_name
from . import metrics

__all__ = ["build_loader_model_grapher", "execute_graph", "train", "test",
           "run", "Grapher", "CheckpointBundle", "ModelSaver", "get_name",
           "metrics"]
