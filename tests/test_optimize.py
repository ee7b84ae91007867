"""L-BFGS-B wrapper behavior: bounds, convergence, and the
restart-on-bound-collapse guard (the Airfoil regression of round 1: a
collapsed beta -> constant-kernel local optimum; see optimize.py).
"""

import numpy as np

from spark_gp_amd.optimize import MemoizedObjective, lbfgsb


def quad_factory(center):
    def fn(x):
        d = x - center
        return float(d @ d), 2.0 * d
    return fn


def test_lbfgsb_unconstrained_quadratic():
    x = lbfgsb(quad_factory(np.array([1.5, -2.0])), np.zeros(2),
               np.full(2, -10.0), np.full(2, 10.0))
    np.testing.assert_allclose(x, [1.5, -2.0], atol=1e-5)


def test_lbfgsb_respects_bounds():
    x = lbfgsb(quad_factory(np.array([5.0])), np.zeros(1),
               np.array([-1.0]), np.array([2.0]))
    np.testing.assert_allclose(x, [2.0], atol=1e-6)


def test_restart_on_bound_collapse_escapes_boundary_optimum():
    """A function with a poor local optimum pinned at the lower bound and a
    better interior optimum: the restart guard must find the interior one
    when the first solve collapses."""
    calls = [0]

    def fn(x):
        calls[0] += 1
        v = float(x[0])
        # narrow collapse basin at the lower bound (value 1), global min at
        # v=2 (value 0); the restart nudge (lo + 0.01*(x0-lo or 1)) lands
        # outside the basin, like the real beta-collapse geometry
        if v < 0.005:
            return 1.0 + v, np.array([1.0])
        return (v - 2.0) ** 2, np.array([2.0 * (v - 2.0)])

    lo, hi = np.array([0.0]), np.array([10.0])
    x0 = np.array([0.003])         # rolls down to the bound
    x_norestart = lbfgsb(fn, x0, lo, hi, restart_on_bound_collapse=False)
    assert x_norestart[0] < 0.005  # stuck at the boundary optimum
    x_restart = lbfgsb(fn, x0, lo, hi, restart_on_bound_collapse=True)
    np.testing.assert_allclose(x_restart, [2.0], atol=1e-4)


def test_restart_keeps_better_boundary_solution():
    """If the boundary solution IS the optimum, the guard must not make
    things worse."""
    fn = quad_factory(np.array([-5.0]))     # optimum below the lower bound
    x = lbfgsb(fn, np.array([1.0]), np.array([0.0]), np.array([10.0]),
               restart_on_bound_collapse=True)
    np.testing.assert_allclose(x, [0.0], atol=1e-6)


def test_memoized_objective_separate_points():
    ev = [0]

    def fn(x):
        ev[0] += 1
        return float(x @ x), 2 * x

    mo = MemoizedObjective(fn)
    a = np.array([1.0, 2.0])
    f1, g1 = mo(a)
    f2, g2 = mo(a.copy())          # same point, different array object
    assert ev[0] == 1
    assert f1 == f2 and np.array_equal(g1, g2)
    mo(np.array([3.0, 4.0]))
    assert ev[0] == 2
