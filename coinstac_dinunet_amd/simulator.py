"""In-process loopback cluster: emulates the COINSTAC file/JSON relay.

The reference depends on an EXTERNAL Node.js simulator to shuttle
transferDirectory files and output->input JSON between N local sites and
the remote. This module is that relay, in-process, so the
full COINNLocal/COINNRemote lock-step protocol runs on one machine with no
platform: used by the CPU plumbing tests and as a standalone debugging
harness for user computations.

Round semantics (matching the simulator):
  1. every site runs COINNLocal.compute(input=last remote output);
  2. each site's transferDirectory is moved into the remote's
     baseDirectory/<site>/ and its output dict becomes remote input[<site>];
  3. the remote runs COINNRemote.compute();
  4. the remote's transferDirectory is moved into every site's
     baseDirectory and its output dict becomes every site's next input.
"""
import os
import shutil


class SiteState:
    def __init__(self, root, client_id):
        self.clientId = client_id
        self.baseDirectory = os.path.join(root, client_id, 'input')
        self.transferDirectory = os.path.join(root, client_id, 'transfer')
        self.outputDirectory = os.path.join(root, client_id, 'output')
        self.cacheDirectory = os.path.join(root, client_id, 'cache')
        for d in (self.baseDirectory, self.transferDirectory,
                  self.outputDirectory, self.cacheDirectory):
            os.makedirs(d, exist_ok=True)

    def as_dict(self):
        return {'clientId': self.clientId,
                'baseDirectory': self.baseDirectory,
                'transferDirectory': self.transferDirectory,
                'outputDirectory': self.outputDirectory,
                'cacheDirectory': self.cacheDirectory}


def _move_contents(src_dir, dst_dir):
    os.makedirs(dst_dir, exist_ok=True)
    for name in os.listdir(src_dir):
        src = os.path.join(src_dir, name)
        dst = os.path.join(dst_dir, name)
        if os.path.isdir(dst):
            shutil.rmtree(dst)
        elif os.path.exists(dst):
            os.remove(dst)
        shutil.move(src, dst)


class LoopbackCluster:
    """Drive N local sites + 1 remote through the full phase machine."""

    def __init__(self, root, n_sites=2, site_data=None):
        """site_data: optional callable(site_state) to populate each site's
        baseDirectory before the first round."""
        self.root = root
        self.sites = [SiteState(root, f'local{i}') for i in range(n_sites)]
        self.remote_state = SiteState(root, 'remote')
        self.site_caches = [{} for _ in range(n_sites)]
        self.remote_cache = {}
        self.site_inputs = [{} for _ in range(n_sites)]
        self.remote_input = {}
        self.rounds = 0
        if site_data:
            for s in self.sites:
                site_data(s)

    def run(self, make_local, make_remote, trainer_cls, dataset_cls=None,
            mp_pool=None, max_rounds=500, **compute_kw):
        """make_local(cache, input, state) -> COINNLocal;
        make_remote(cache, input, state) -> COINNRemote.
        Returns (success, remote_output)."""
        success, remote_out = False, {}
        reducer_cls = compute_kw.pop('reducer_cls', None)
        remote_kw = {'reducer_cls': reducer_cls} if reducer_cls else {}
        for self.rounds in range(1, max_rounds + 1):
            # 1) site computations
            site_outs = {}
            for i, site in enumerate(self.sites):
                local = make_local(self.site_caches[i],
                                   dict(self.site_inputs[i]), site.as_dict())
                result = local(mp_pool, trainer_cls, dataset_cls=dataset_cls,
                               **compute_kw)
                site_outs[site.clientId] = result['output']
            # 2) site transfers -> remote inbox
            for site in self.sites:
                _move_contents(site.transferDirectory,
                               os.path.join(self.remote_state.baseDirectory,
                                            site.clientId))
            # 3) remote computation
            remote = make_remote(self.remote_cache, site_outs,
                                 self.remote_state.as_dict())
            rres = remote(mp_pool, trainer_cls, **remote_kw)
            remote_out, success = rres['output'], rres.get('success', False)
            if success:
                break
            # 4) remote transfer -> every site's inbox; remote out -> inputs
            for site in self.sites:
                for name in os.listdir(self.remote_state.transferDirectory):
                    src = os.path.join(self.remote_state.transferDirectory, name)
                    dst = os.path.join(site.baseDirectory, name)
                    if os.path.isfile(src):
                        shutil.copy(src, dst)
                    else:
                        if os.path.isdir(dst):
                            shutil.rmtree(dst)
                        shutil.copytree(src, dst)
            for name in os.listdir(self.remote_state.transferDirectory):
                p = os.path.join(self.remote_state.transferDirectory, name)
                shutil.rmtree(p) if os.path.isdir(p) else os.remove(p)
            self.site_inputs = [dict(remote_out) for _ in self.sites]
        return success, remote_out
