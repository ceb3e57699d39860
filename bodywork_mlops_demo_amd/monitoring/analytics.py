"""Model-performance analytics over the metric history.

Capability parity with the reference's
``notebooks/model-performance-analytics.ipynb`` (C9):
download **all** ``model-metrics/`` and ``test-metrics/`` CSVs, concat
each family into a time-indexed DataFrame, and summarise drift — how the
live (online) MAPE of the deployed model degrades on next-period data
relative to its offline training MAPE.
"""
from __future__ import annotations

import argparse

import pandas as pd

from bodywork_mlops_demo_amd.store import ArtefactStore, contract, open_store
from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)


def download_metrics(store: ArtefactStore, prefix: str) -> pd.DataFrame:
    """All metric CSVs under a prefix → one time-ordered DataFrame
    (the notebook's ``download_metrics`` capability)."""
    rows = []
    for key, d in store.all_by_date(prefix):
        rec = store.get_metrics_csv(key)
        rec["date"] = str(d)
        rows.append(rec)
    if not rows:
        return pd.DataFrame()
    df = pd.DataFrame(rows)
    for col in df.columns:
        if col not in ("date", "response_time_kind"):
            df[col] = pd.to_numeric(df[col], errors="coerce")
    df["date"] = pd.to_datetime(df["date"])
    return df.sort_values("date").reset_index(drop=True)


def drift_report(store: ArtefactStore) -> dict:
    """Joined offline/online metric history + drift summary statistics."""
    offline = download_metrics(store, contract.MODEL_METRICS_PREFIX)
    online = download_metrics(store, contract.TEST_METRICS_PREFIX)
    report: dict = {"offline": offline, "online": online}
    if not offline.empty and not online.empty:
        joined = offline.merge(
            online, on="date", suffixes=("_offline", "_online")
        )
        report["joined"] = joined
        report["summary"] = {
            "days": len(joined),
            "mean_offline_MAPE": float(joined["MAPE_offline"].mean()),
            "mean_online_MAPE": float(joined["MAPE_online"].mean()),
            "max_online_MAPE": float(joined["MAPE_online"].max()),
            "mean_drift_gap": float(
                (joined["MAPE_online"] - joined["MAPE_offline"]).mean()
            ),
            "mean_response_time": float(
                joined["mean_response_time"].mean()
            ) if "mean_response_time" in joined else None,
        }
    return report


def plot_drift(report: dict, path: str) -> str:
    """Render the offline-vs-online MAPE history and the drift gap to a
    PNG (the reference analytics notebook's seaborn plots as a durable
    CLI artefact — `model-performance-analytics.ipynb` capability)."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    j = report["joined"]
    fig, (ax1, ax2) = plt.subplots(2, 1, figsize=(10, 6), sharex=True)
    ax1.plot(j["date"], j["MAPE_offline"], "o-",
             label="offline MAPE (train-time)")
    ax1.plot(j["date"], j["MAPE_online"], "s-",
             label="online MAPE (live service, t+1 data)")
    ax1.set_yscale("log")
    ax1.legend()
    ax1.grid(alpha=0.3)
    ax1.set_title("model quality across pipeline days")
    ax2.plot(j["date"], j["MAPE_online"] - j["MAPE_offline"], "k.-")
    ax2.axhline(0, color="gray", lw=0.5)
    ax2.grid(alpha=0.3)
    ax2.set_title("drift gap: online - offline MAPE")
    fig.autofmt_xdate()
    fig.savefig(path, dpi=90, bbox_inches="tight")
    plt.close(fig)
    return path


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--store", default=None)
    p.add_argument("--csv-out", default=None,
                   help="write the joined history as CSV")
    p.add_argument("--plot", default=None, metavar="PNG",
                   help="render the drift history to a PNG")
    args = p.parse_args(argv)
    report = drift_report(open_store(args.store))
    if "joined" in report:
        pd.set_option("display.width", 160)
        print(report["joined"].to_string(index=False))
        print("\nsummary:", report["summary"])
        if args.csv_out:
            report["joined"].to_csv(args.csv_out, index=False)
        if args.plot:
            print("wrote", plot_drift(report, args.plot))
    else:
        print("no joint metric history yet "
              f"(offline={len(report['offline'])}, online={len(report['online'])})")


if __name__ == "__main__":
    main()
