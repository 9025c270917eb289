/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package bookstore

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	appsv1alpha1 "github.com/acme/bookstore/apis/apps/v1alpha1"
)

// +kubebuilder:rbac:groups=apps,resources=deployments,verbs=get;list;watch;create;update;patch;delete

const DeploymentBookstoreDeploy = "bookstore-deploy"

// CreateDeploymentBookstoreDeploy creates the bookstore-deploy Deployment resource.
func CreateDeploymentBookstoreDeploy(
	parent *appsv1alpha1.BookStore,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apps/v1",
			"kind": "Deployment",
			"metadata": map[string]interface{}{
				"name": "bookstore-deploy",
			},
			"spec": map[string]interface{}{
				"replicas": parent.Spec.Bookstore.Deeply.Nested.Path.Replicas,
				"selector": map[string]interface{}{
					"matchLabels": map[string]interface{}{
						"app": parent.Spec.App.Label,
					},
				},
				"template": map[string]interface{}{
					"metadata": map[string]interface{}{
						"labels": map[string]interface{}{
							"app": "bookstore",
						},
					},
					"spec": map[string]interface{}{
						"containers": []interface{}{
							map[string]interface{}{
								"name": "bookstore-container",
								"image": parent.Spec.Bookstore.Deeply.Nested.Image,
								"ports": []interface{}{
									map[string]interface{}{
										"containerPort": 8080,
									},
								},
							},
						},
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=networking.k8s.io,resources=ingresses,verbs=get;list;watch;create;update;patch;delete

const IngressBookstoreIng = "bookstore-ing"

// CreateIngressBookstoreIng creates the bookstore-ing Ingress resource.
func CreateIngressBookstoreIng(
	parent *appsv1alpha1.BookStore,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "networking.k8s.io/v1",
			"kind": "Ingress",
			"metadata": map[string]interface{}{
				"name": "bookstore-ing",
				"annotations": map[string]interface{}{
					"nginx.ingress.kubernetes.io/rewrite-target": "/",
				},
			},
			"spec": map[string]interface{}{
				"rules": []interface{}{
					map[string]interface{}{
						"host": "shop.example.com",
						"http": map[string]interface{}{
							"paths": []interface{}{
								map[string]interface{}{
									"path": "/",
									"pathType": "Prefix",
									"backend": map[string]interface{}{
										"service": map[string]interface{}{
											"name": "bookstore-svc",
											"port": map[string]interface{}{
												"number": 80,
											},
										},
									},
								},
							},
						},
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=core,resources=services,verbs=get;list;watch;create;update;patch;delete

// CreateServiceServiceNameSvc creates the !!start parent.Spec.Service.Name !!end-svc Service resource.
func CreateServiceServiceNameSvc(
	parent *appsv1alpha1.BookStore,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"kind": "Service",
			"apiVersion": "v1",
			"metadata": map[string]interface{}{
				"name": parent.Spec.Service.Name + "-svc",
			},
			"spec": map[string]interface{}{
				"selector": map[string]interface{}{
					"app": parent.Spec.App.Label,
				},
				"ports": []interface{}{
					map[string]interface{}{
						"protocol": "TCP",
						"port": parent.Spec.Bookstore.Deeply.Nested.Path.Required.Port,
						"targetPort": parent.Spec.Service.TargetPort,
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=rbac.authorization.k8s.io,resources=roles,verbs=get;list;watch;create;update;patch;delete
// +kubebuilder:rbac:groups=apps,resources=pods,verbs=get;list;watch;create;update;delete
// +kubebuilder:rbac:groups=apps,resources=deployments,verbs=get;list;watch;create;update;delete
// +kubebuilder:rbac:groups=core,resources=secrets,verbs=get;list;watch;create;update;delete
// +kubebuilder:rbac:groups=core,resources=events,verbs=create;patch

const RoleBookstoreRole = "bookstore-role"

// CreateRoleBookstoreRole creates the bookstore-role Role resource.
func CreateRoleBookstoreRole(
	parent *appsv1alpha1.BookStore,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "rbac.authorization.k8s.io/v1",
			"kind": "Role",
			"metadata": map[string]interface{}{
				"name": "bookstore-role",
			},
			"rules": []interface{}{
				map[string]interface{}{
					"apiGroups": []interface{}{
						"apps",
					},
					"resources": []interface{}{
						"pods",
						"deployments",
					},
					"verbs": []interface{}{
						"get",
						"list",
						"watch",
						"create",
						"update",
						"delete",
					},
				},
				map[string]interface{}{
					"apiGroups": []interface{}{
						"",
					},
					"resources": []interface{}{
						"secrets",
					},
					"verbs": []interface{}{
						"get",
						"list",
						"watch",
						"create",
						"update",
						"delete",
					},
				},
				map[string]interface{}{
					"apiGroups": []interface{}{
						"",
					},
					"resources": []interface{}{
						"events",
					},
					"verbs": []interface{}{
						"create",
						"patch",
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}
