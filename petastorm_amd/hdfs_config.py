"""HDFS HA namenode resolution from Hadoop site configuration.

Parity: /root/reference/petastorm/hdfs/namenode.py:31-128
(HdfsNamenodeResolver).  The reference resolves a nameservice to its
namenode list and hands them to a libhdfs3 connector with per-call
failover (namenode.py:146-238); this framework standardizes on fsspec
(fs_utils docstring), so resolution produces fsspec ``storage_options``
(host/port candidates) and transient-failure retries are handled
generically by :class:`petastorm_amd.fs_utils.RetryingFilesystem`.

Pure host-side config parsing — fully testable without a Hadoop cluster
(the reference itself only ever tests this against mock configurations,
hdfs/tests/test_hdfs_namenode.py).
"""

import logging
import os
import xml.etree.ElementTree as ET
from urllib.parse import urlparse

logger = logging.getLogger(__name__)

#: Environment variables naming the Hadoop install, in the reference's
#: preference order (namenode.py:45).
HADOOP_HOME_ENVS = ('HADOOP_HOME', 'HADOOP_PREFIX', 'HADOOP_INSTALL')


def load_hadoop_configuration(hadoop_path=None):
    """Parse hdfs-site.xml + core-site.xml into a flat {name: value} dict
    (reference namenode.py:68-75).  ``hadoop_path`` defaults to the first
    of HADOOP_HOME/HADOOP_PREFIX/HADOOP_INSTALL present in the
    environment; returns {} when nothing is configured."""
    if hadoop_path is None:
        for env in HADOOP_HOME_ENVS:
            if env in os.environ:
                hadoop_path = os.environ[env]
                break
    conf = {}
    if hadoop_path is None:
        logger.warning(
            'No Hadoop configuration found (set HADOOP_HOME to enable '
            'HDFS nameservice resolution)')
        return conf
    for fname in ('hdfs-site.xml', 'core-site.xml'):
        path = os.path.join(hadoop_path, 'etc', 'hadoop', fname)
        if not os.path.exists(path):
            continue
        try:
            for prop in ET.parse(path).getroot().iter('property'):
                name = prop.find('name')
                value = prop.find('value')
                if name is not None and value is not None:
                    conf[name.text] = value.text
        except ET.ParseError as e:
            logger.error('Unparseable Hadoop site file %s: %s', path, e)
    return conf


class HdfsNamenodeResolver(object):
    """Resolve HDFS nameservices to namenode host:port lists
    (reference namenode.py:31-128 semantics, including the error
    behaviors its mock tests pin down)."""

    def __init__(self, hadoop_configuration=None):
        if hadoop_configuration is None:
            hadoop_configuration = load_hadoop_configuration()
        self._conf = hadoop_configuration

    def resolve_hdfs_name_service(self, namespace):
        """Namenode URL list for ``namespace``, or None when the name is
        not a configured nameservice (it may simply be a hostname).
        Raises when the nameservice exists but a namenode address is
        missing (misconfiguration must be loud)."""
        namenodes = self._conf.get('dfs.ha.namenodes.' + namespace)
        if not namenodes:
            return None
        urls = []
        for nn in namenodes.split(','):
            key = 'dfs.namenode.rpc-address.{}.{}'.format(
                namespace, nn.strip())
            url = self._conf.get(key)
            if not url:
                raise RuntimeError(
                    'Hadoop configuration names namenode {!r} for '
                    'nameservice {!r} but has no {} property'
                    .format(nn, namespace, key))
            urls.append(url)
        return urls

    def resolve_default_hdfs_service(self):
        """(nameservice, [namenode urls]) from fs.defaultFS
        (reference namenode.py:112-128)."""
        default_fs = self._conf.get('fs.defaultFS')
        if not default_fs:
            raise RuntimeError(
                'Hadoop configuration has no fs.defaultFS property')
        nameservice = urlparse(default_fs).netloc
        namenodes = self.resolve_hdfs_name_service(nameservice)
        if namenodes is None:
            raise IOError(
                'Unable to resolve namenodes for default nameservice '
                '{!r}'.format(default_fs))
        return nameservice, namenodes


def hdfs_storage_options(url, hadoop_configuration=None):
    """fsspec storage_options for an hdfs:// URL.

    A URL naming a configured HA nameservice resolves to the FIRST
    namenode's host/port plus the full candidate list under
    ``fallback_namenodes`` (RetryingFilesystem retries transient
    failures; fsspec's libhdfs layer performs its own HA failover when
    given the nameservice).  A URL with an explicit host[:port] passes
    through unchanged.
    """
    parsed = urlparse(url)
    netloc = parsed.netloc
    if not netloc:
        resolver = HdfsNamenodeResolver(hadoop_configuration)
        service, namenodes = resolver.resolve_default_hdfs_service()
        host, _, port = namenodes[0].partition(':')
        return {'host': host, 'port': int(port) if port else 8020,
                'fallback_namenodes': namenodes[1:]}
    resolver = HdfsNamenodeResolver(hadoop_configuration)
    namenodes = resolver.resolve_hdfs_name_service(netloc)
    if namenodes:
        host, _, port = namenodes[0].partition(':')
        return {'host': host, 'port': int(port) if port else 8020,
                'fallback_namenodes': namenodes[1:]}
    host, _, port = netloc.partition(':')
    return {'host': host, 'port': int(port) if port else 8020,
            'fallback_namenodes': []}
