"""Fetch pretrained models (reference `distar/bin/download_model.py`).

This image has no network egress; the command therefore resolves models in
this order and copies them into distar_amd/bin/:
  1. a local path given via --path,
  2. $DISTAR_AMD_MODEL_DIR/<name>.pth,
  3. (online deployments) the reference's release URLs via requests.
"""
import argparse
import os
import shutil

MODEL_URLS = {
    'sl_model': 'https://github.com/opendilab/DI-star/releases/download/v0.1.0/sl_model.pth',
    'rl_model': 'https://github.com/opendilab/DI-star/releases/download/v0.1.0/rl_model.pth',
}


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--name', default='rl_model', help='sl_model | rl_model | <named>')
    p.add_argument('--path', default=None, help='local checkpoint to install')
    args = p.parse_args(argv)
    dst = os.path.join(os.path.dirname(__file__), f'{args.name}.pth')
    if args.path and os.path.exists(args.path):
        shutil.copyfile(args.path, dst)
        print(f'installed {args.path} -> {dst}')
        return
    local_dir = os.environ.get('DISTAR_AMD_MODEL_DIR')
    if local_dir:
        src = os.path.join(local_dir, f'{args.name}.pth')
        if os.path.exists(src):
            shutil.copyfile(src, dst)
            print(f'installed {src} -> {dst}')
            return
    url = MODEL_URLS.get(args.name)
    if url is None:
        raise SystemExit(f'unknown model {args.name!r} and no local source given')
    try:
        import requests
        print(f'downloading {url} ...')
        r = requests.get(url, timeout=600)
        r.raise_for_status()
        with open(dst, 'wb') as f:
            f.write(r.content)
        print(f'saved {dst}')
    except Exception as e:  # noqa: BLE001
        raise SystemExit(
            f'no network access ({e!r}); pass --path or set '
            'DISTAR_AMD_MODEL_DIR to install a local checkpoint')


if __name__ == '__main__':
    main()
