from .ring import (
    neighbour_exchange,
    neighbour_exchange_bidir,
    neighbour_exchange_with_grad,
    neighbour_exchange_bidir_with_grad,
    neighbour_exchange_start,
    neighbour_exchange_bidir_start,
    quantized_exchange_start,
    quantized_exchange_bidir_start,
    NeighbourExchange,
    NeighbourExchangeBidir,
    RingHandle,
)
from .collectives import all_gather_with_grad, average_gradients

__all__ = [
    "neighbour_exchange",
    "neighbour_exchange_bidir",
    "neighbour_exchange_with_grad",
    "neighbour_exchange_bidir_with_grad",
    "neighbour_exchange_start",
    "neighbour_exchange_bidir_start",
    "quantized_exchange_start",
    "quantized_exchange_bidir_start",
    "NeighbourExchange",
    "NeighbourExchangeBidir",
    "RingHandle",
    "all_gather_with_grad",
    "average_gradients",
]
