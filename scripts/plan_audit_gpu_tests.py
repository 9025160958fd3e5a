"""Plan-audit the GPU test suites on CPU: run every `-m gpu` test function
with a Context whose .sql() PLANS the query and then aborts the test. Each
test therefore exercises its create_table calls and the planning of its
FIRST query (most have one) with no GPU — a regression net for planner
changes made while the GPU pool is closed.
Usage: python scripts/plan_audit_gpu_tests.py"""
import inspect
import sys

sys.path.insert(0, "/root/repo")


class _PlanOK(Exception):
    pass


def patch_context():
    """Class-level patch so tests that build their own Context are audited
    too."""
    from dask_sql_amd.context import Context

    def sql(self, q, **kw):
        up = q.upper().lstrip()
        if up.startswith(("SHOW", "DROP", "ANALYZE", "CREATE")):
            raise _PlanOK()  # statement paths are CPU-tested elsewhere
        self._get_ral(q)
        raise _PlanOK()

    Context.sql = sql


def make_ctx():
    from dask_sql_amd.context import Context
    return Context()


class _MP:
    """Minimal monkeypatch stand-in (setenv/delenv/setattr)."""

    def __init__(self):
        import os
        self._os = os
        self._undo = []

    def setenv(self, k, v):
        old = self._os.environ.get(k)
        self._undo.append((k, old))
        self._os.environ[k] = v

    def delenv(self, k, raising=True):
        old = self._os.environ.pop(k, None)
        self._undo.append((k, old))

    def setattr(self, obj, name, value):
        self._undo.append((obj, name, getattr(obj, name)))
        setattr(obj, name, value)

    def undo(self):
        for item in reversed(self._undo):
            if len(item) == 2:
                k, old = item
                if old is None:
                    self._os.environ.pop(k, None)
                else:
                    self._os.environ[k] = old
            else:
                obj, name, old = item
                setattr(obj, name, old)


def run_module(modname, fixture_makers):
    import importlib
    m = importlib.import_module(modname)
    ok = aborted = failed = 0
    for name in sorted(n for n in dir(m) if n.startswith("test_")):
        fn = getattr(m, name)
        params = list(inspect.signature(fn).parameters)
        ctx = make_ctx()
        args = []
        skip = False
        for p in params:
            if p == "c" and "c" in fixture_makers:
                try:
                    args.append(fixture_makers["c"](ctx))
                except BaseException:
                    skip = True
                    break
            elif p == "ctx" or p == "c":
                args.append(ctx)
            elif p == "monkeypatch":
                args.append(_MP())
            elif p == "tmp_path":
                import pathlib
                import tempfile
                args.append(pathlib.Path(tempfile.mkdtemp()))
            elif p in ("seed", "qi"):
                args.append(0)
            elif p in fixture_makers:
                try:
                    mk = fixture_makers[p]
                    nargs = len(inspect.signature(mk).parameters)
                    args.append(mk(ctx) if nargs else mk())
                except _PlanOK:
                    skip = True
                    break
                except BaseException:
                    skip = True  # fixture itself skips without a GPU
                    break
            else:
                skip = True
                break
        if skip:
            aborted += 1
            continue
        try:
            fn(*args)
            ok += 1  # no sql issued (pure create_table etc.)
        except _PlanOK:
            ok += 1
        except Exception as e:
            failed += 1
            print(f"PLAN-FAIL {modname}.{name}: {type(e).__name__}: "
                  f"{str(e)[:110]}")
        finally:
            for a_ in args:
                if isinstance(a_, _MP):
                    a_.undo()
    return ok, aborted, failed


def main():
    patch_context()
    # fixtures gate on torch.cuda.is_available() before any device work;
    # planning never touches the device, so let them through
    import torch
    torch.cuda.is_available = lambda: True
    total_ok = total_ab = total_fail = 0
    for modname in ("tests.test_gpu_parity", "tests.test_gpu_semantics",
                    "tests.test_gpu_tpch_mini", "tests.test_zz_r2_surface",
                    "tests.test_zz_sqlite_compat"):
        import importlib
        m = importlib.import_module(modname)
        makers = {}
        try:
            from tests import conftest as _cf
            for fx in ("user_table_1", "user_table_2", "df_simple"):
                f = getattr(_cf, fx, None)
                if f is not None and hasattr(f, "__wrapped__"):
                    makers[fx] = f.__wrapped__
        except Exception:
            pass
        cfx = getattr(m, "c", None)
        if cfx is not None and hasattr(cfx, "__wrapped__"):
            makers["c"] = cfx.__wrapped__
        for fx in ("tpch", "tpch2"):
            f = getattr(m, fx, None)
            if f is not None and hasattr(f, "__wrapped__"):
                makers[fx] = f.__wrapped__
        ok, ab, fail = run_module(modname, makers)
        total_ok += ok
        total_ab += ab
        total_fail += fail
        print(f"{modname}: planned-ok {ok}, not-auditable {ab}, "
              f"FAILED {fail}")
    print(f"\nTOTAL planned-ok {total_ok}, not-auditable {total_ab}, "
          f"FAILED {total_fail}")
    return 1 if total_fail else 0


if __name__ == "__main__":
    sys.exit(main())
