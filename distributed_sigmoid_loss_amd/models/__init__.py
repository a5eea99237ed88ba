from .towers import make_toy_towers, ProjectionTower, TwoTowerModel

__all__ = ["make_toy_towers", "ProjectionTower", "TwoTowerModel"]
