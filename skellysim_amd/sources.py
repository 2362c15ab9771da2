"""Point sources and background source flows — the reference's
PointSourceContainer (src/core/point_source.cpp) and BackgroundSource
(src/core/background_source.cpp): static point forces/torques with optional
lifetimes (regularized-Stokeslet + rotlet flow) and an affine background
velocity field. Both enter the solver RHS through prep
(system.cpp:445-446) and the post-processing velocity field
(system.cpp:357-359)."""

import numpy as np


class PointSource:
    """One point forcer/torquer (point_source.hpp:7-15); time_to_live 0
    means always alive (point_source.cpp:22-24)."""

    def __init__(self, position=(0.0, 0.0, 0.0), force=(0.0, 0.0, 0.0),
                 torque=(0.0, 0.0, 0.0), time_to_live=0.0):
        self.position = np.asarray(position, float)
        self.force = np.asarray(force, float)
        self.torque = np.asarray(torque, float)
        self.time_to_live = float(time_to_live)


class PointSourceContainer:
    """point_source.cpp:16-55: regularized-Stokeslet (oseen contract) flow
    of the live forces + rotlet flow of the live torques."""

    def __init__(self, points=()):
        self.points = list(points)

    @classmethod
    def from_config(cls, tables):
        return cls([PointSource(position=t.get("position", (0, 0, 0)),
                                force=t.get("force", (0, 0, 0)),
                                torque=t.get("torque", (0, 0, 0)),
                                time_to_live=t.get("time_to_live", 0.0))
                    for t in tables])

    def flow(self, r_trg, eta, time, backend):
        vel = np.zeros_like(np.asarray(r_trg, float))
        live = [p for p in self.points
                if not (p.time_to_live and time >= p.time_to_live)]
        forcers = [p for p in live if p.force.any()]
        torquers = [p for p in live if p.torque.any()]
        if forcers:
            vel += backend.oseen_contract(
                np.stack([p.position for p in forcers]),
                np.asarray(r_trg, float),
                np.stack([p.force for p in forcers]), eta)
        if torquers:
            vel += backend.rotlet(
                np.stack([p.position for p in torquers]),
                np.stack([p.torque for p in torquers]),
                np.asarray(r_trg, float), eta)
        return vel


class BackgroundSource:
    """background_source.cpp:14-22: v_j(r) = uniform_j +
    r[components_j] * scale_factor_j."""

    def __init__(self, components=(0, 1, 2), scale_factor=(0.0, 0.0, 0.0),
                 uniform=(0.0, 0.0, 0.0)):
        self.components = np.asarray(components, int)
        self.scale_factor = np.asarray(scale_factor, float)
        self.uniform = np.asarray(uniform, float)

    @classmethod
    def from_config(cls, table):
        return cls(components=table.get("components", (0, 1, 2)),
                   scale_factor=table.get("scale_factor", (0.0, 0.0, 0.0)),
                   uniform=table.get("uniform", (0.0, 0.0, 0.0)))

    def is_active(self):
        return bool(np.linalg.norm(self.uniform)
                    + np.linalg.norm(self.scale_factor))

    def flow(self, r_trg, eta=None):
        r = np.asarray(r_trg, float)
        return self.uniform[None, :] + r[:, self.components] \
            * self.scale_factor[None, :]
