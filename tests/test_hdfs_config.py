"""HDFS HA namenode resolution against on-disk Hadoop site files
(reference hdfs/tests/test_hdfs_namenode.py tests the same semantics
against mock configurations)."""
import os

import pytest

from petastorm_amd.hdfs_config import (HdfsNamenodeResolver,
                                       hdfs_storage_options,
                                       load_hadoop_configuration)

HDFS_SITE = """<?xml version="1.0"?>
<configuration>
  <property><name>dfs.ha.namenodes.mycluster</name>
            <value>nn1,nn2</value></property>
  <property><name>dfs.namenode.rpc-address.mycluster.nn1</name>
            <value>namenode-a.example.com:8020</value></property>
  <property><name>dfs.namenode.rpc-address.mycluster.nn2</name>
            <value>namenode-b.example.com:8020</value></property>
  <property><name>dfs.ha.namenodes.broken</name>
            <value>nn1</value></property>
</configuration>
"""

CORE_SITE = """<?xml version="1.0"?>
<configuration>
  <property><name>fs.defaultFS</name>
            <value>hdfs://mycluster</value></property>
</configuration>
"""


@pytest.fixture()
def hadoop_home(tmp_path, monkeypatch):
    conf_dir = tmp_path / 'etc' / 'hadoop'
    conf_dir.mkdir(parents=True)
    (conf_dir / 'hdfs-site.xml').write_text(HDFS_SITE)
    (conf_dir / 'core-site.xml').write_text(CORE_SITE)
    monkeypatch.setenv('HADOOP_HOME', str(tmp_path))
    for env in ('HADOOP_PREFIX', 'HADOOP_INSTALL'):
        monkeypatch.delenv(env, raising=False)
    return str(tmp_path)


def test_load_configuration_merges_both_files(hadoop_home):
    conf = load_hadoop_configuration()
    assert conf['fs.defaultFS'] == 'hdfs://mycluster'
    assert conf['dfs.ha.namenodes.mycluster'] == 'nn1,nn2'


def test_resolve_nameservice(hadoop_home):
    r = HdfsNamenodeResolver()
    assert r.resolve_hdfs_name_service('mycluster') == [
        'namenode-a.example.com:8020', 'namenode-b.example.com:8020']


def test_resolve_plain_hostname_returns_none(hadoop_home):
    assert HdfsNamenodeResolver().resolve_hdfs_name_service(
        'some-host.example.com') is None


def test_missing_rpc_address_raises(hadoop_home):
    # nameservice 'broken' lists nn1 but has no rpc-address property
    with pytest.raises(RuntimeError, match='rpc-address'):
        HdfsNamenodeResolver().resolve_hdfs_name_service('broken')


def test_resolve_default_service(hadoop_home):
    service, namenodes = HdfsNamenodeResolver() \
        .resolve_default_hdfs_service()
    assert service == 'mycluster'
    assert len(namenodes) == 2


def test_default_service_without_config_raises(monkeypatch, tmp_path):
    for env in ('HADOOP_HOME', 'HADOOP_PREFIX', 'HADOOP_INSTALL'):
        monkeypatch.delenv(env, raising=False)
    with pytest.raises(RuntimeError, match='fs.defaultFS'):
        HdfsNamenodeResolver({}).resolve_default_hdfs_service()


def test_storage_options_for_nameservice(hadoop_home):
    opts = hdfs_storage_options('hdfs://mycluster/path/to/ds')
    assert opts['host'] == 'namenode-a.example.com'
    assert opts['port'] == 8020
    assert opts['fallback_namenodes'] == ['namenode-b.example.com:8020']


def test_storage_options_for_explicit_host(hadoop_home):
    opts = hdfs_storage_options('hdfs://other-nn:9000/path')
    assert opts['host'] == 'other-nn' and opts['port'] == 9000
    assert opts['fallback_namenodes'] == []


def test_storage_options_bare_default(hadoop_home):
    opts = hdfs_storage_options('hdfs:///path/only')
    assert opts['host'] == 'namenode-a.example.com'
