"""tfprof-style per-op profiling summary from StepStats (compact analog of
reference tools/tfprof/: aggregates node timings by op type / name scope)."""


def profile(step_stats, group_by='op', top=20):
    """Returns a list of dicts {name, count, total_us, avg_us} sorted by
    total time; group_by is 'op', 'node', or 'scope'."""
    agg = {}
    for ds in step_stats.dev_stats:
        for ns in ds.node_stats:
            if group_by == 'op':
                key = ns.op
            elif group_by == 'scope':
                key = ns.node_name.split('/')[0]
            else:
                key = ns.node_name
            a = agg.setdefault(key, [0, 0])
            a[0] += 1
            a[1] += ns.op_end_rel_micros
    rows = [{'name': k, 'count': c, 'total_us': t,
             'avg_us': t / c if c else 0.0}
            for k, (c, t) in agg.items()]
    rows.sort(key=lambda r: -r['total_us'])
    return rows[:top]


def print_profile(step_stats, group_by='op', top=20, file=None):
    rows = profile(step_stats, group_by, top)
    total = sum(r['total_us'] for r in rows) or 1
    print('%-40s %8s %12s %10s %6s' % (group_by, 'count', 'total_us',
                                       'avg_us', '%'), file=file)
    for r in rows:
        print('%-40s %8d %12.0f %10.1f %6.1f' %
              (r['name'][:40], r['count'], r['total_us'], r['avg_us'],
               100.0 * r['total_us'] / total), file=file)
    return rows
