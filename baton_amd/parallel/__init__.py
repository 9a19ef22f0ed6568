from baton_amd.parallel.data_plane import FederatedDataPlane

__all__ = ["FederatedDataPlane"]
