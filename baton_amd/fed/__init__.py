from baton_amd.fed.aggregate import fedavg_, weighted_loss_history
from baton_amd.fed.split import dirichlet_partition, iid_partition
from baton_amd.fed.dataset import FederatedTensorDataset

__all__ = ["fedavg_", "weighted_loss_history", "dirichlet_partition", "iid_partition", "FederatedTensorDataset"]
