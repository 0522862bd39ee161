from ding.utils.k8s_helper import OperatorServer
