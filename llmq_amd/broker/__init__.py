from llmq_amd.broker.server import BrokerServer

__all__ = ["BrokerServer"]
