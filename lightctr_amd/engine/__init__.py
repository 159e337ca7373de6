from .dag import (ActivationOp, AddOp, AggregateNode, DAGPipeline, LossOp,
                  MatmulOp, MultiplyOp, Node, SourceNode, TerminusNode,
                  TrainableNode)

__all__ = ["Node", "SourceNode", "TrainableNode", "TerminusNode",
           "AggregateNode", "AddOp", "MultiplyOp", "MatmulOp",
           "ActivationOp", "LossOp", "DAGPipeline"]
