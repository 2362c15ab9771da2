"""Dynamic instability — nucleate/grow/remove fibers at body nucleation
sites (the reference's System::dynamic_instability,
src/core/dynamic_instability.cpp:25-194, single-rank form).

Per solver prep (called first, system.cpp:403):
  - every fiber gets v_growth = params.v_growth and catastrophe frequency
    f_catastrophe, both collision-scaled when its plus end is pinned at the
    cortex (lines 73-78);
  - each fiber is removed with probability 1 - exp(-dt * f_cat) (line 83);
  - survivors grow: length_prev = length, length += dt * v_growth;
  - Poisson(dt * nucleation_rate * n_inactive_sites) new fibers nucleate at
    randomly chosen unoccupied body sites, pointing radially outward with
    length min_length and v_growth 0 (lines 104-193).

RNG: the reference uses its own splittable Philox streams (rng.cpp); this
engine uses one numpy Generator owned by the system (same distributions,
different stream — a documented deviation).
"""

import numpy as np

from .fiber_fd import FiberFD, BC_VELOCITY

DEFAULTS = dict(  # skelly_config.py:312-349
    n_nodes=0, v_growth=0.0, f_catastrophe=0.0,
    v_grow_collision_scale=0.5, f_catastrophe_collision_scale=2.0,
    nucleation_rate=0.0, radius=0.025, min_length=0.5,
    bending_rigidity=2.5e-3, min_separation=0.1)


def dynamic_instability(system, params, rng):
    """Mutates system.fibers. params: dict with the DEFAULTS keys; returns
    dict(n_removed, n_nucleated)."""
    p = {**DEFAULTS, **params}
    if p["n_nodes"] == 0:
        return dict(n_removed=0, n_nucleated=0)
    dt = system.dt
    bodies = system.bodies

    occupied = set()
    survivors = []
    n_active_old = 0
    n_removed = 0
    for fib in system.fibers:
        fib.v_growth = p["v_growth"]
        f_cat = p["f_catastrophe"]
        if fib.bc_plus[0] == BC_VELOCITY:  # is_plus_pinned, f_f_d.hpp:158
            fib.v_growth *= p["v_grow_collision_scale"]
            f_cat *= p["f_catastrophe_collision_scale"]
        attached = fib.binding_site[0] >= 0
        if attached:
            n_active_old += 1
        if rng.uniform() > np.exp(-dt * f_cat):
            n_removed += 1
            continue
        if attached:
            occupied.add(tuple(fib.binding_site))
        fib.length_prev = fib.length
        fib.length += dt * fib.v_growth
        survivors.append(fib)
    system.fibers = survivors

    all_sites = [(ib, js) for ib, b in enumerate(bodies)
                 for js in range(len(b.nucleation_sites))]
    inactive = [s for s in all_sites if s not in occupied]
    n_inactive_old = len(all_sites) - n_active_old
    n_new = min(int(rng.poisson(dt * p["nucleation_rate"]
                                * max(0, n_inactive_old))), len(inactive))
    picks = [inactive[i] for i in
             rng.choice(len(inactive), size=n_new, replace=False)] \
        if n_new else []
    for i_body, i_site in picks:
        body = bodies[i_body]
        origin = body.nucleation_sites[i_site]
        u = origin - body.position
        u = u / np.linalg.norm(u)
        s = np.linspace(0.0, p["min_length"], p["n_nodes"])
        x = origin[None, :] + s[:, None] * u[None, :]
        fib = FiberFD(x, length=p["min_length"],
                      bending_rigidity=p["bending_rigidity"],
                      eta=system.eta, radius=p["radius"], minus_clamped=True)
        fib.v_growth = 0.0
        fib.binding_site = (i_body, i_site)
        system.fibers.append(fib)
    return dict(n_removed=n_removed, n_nucleated=len(picks))
