"""Test-support stage: report which version of a package this stage's
interpreter resolves (used by the per-stage env-isolation tests to prove
two stages ran against different pinned versions, mirroring the
reference's divergent per-stage pip lists, bodywork.yaml:14 vs :33)."""
import argparse
import importlib.metadata


def main(argv=None) -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--package", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--tag", default="")
    args = p.parse_args(argv)
    version = importlib.metadata.version(args.package)
    with open(args.out, "a") as f:
        f.write(f"{args.tag}={version}\n")
    print(f"{args.tag}: {args.package}=={version}")


if __name__ == "__main__":
    main()
